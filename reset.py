"""Delete generated node_data artifacts (parity: reference reset.py)."""
import shutil
from pathlib import Path

if __name__ == "__main__":
    base = Path("node_data")
    for child in base.iterdir() if base.exists() else []:
        if child.is_dir():
            shutil.rmtree(child)
            print("removed", child)
