"""Shared example helpers: CLI, synthetic datasets (no network in this
environment — the reference's downloads are replaced by synthetic data of
identical shape; sklearn-digits ships with sklearn and stays real)."""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def node_name(default="node_0"):
    ap = argparse.ArgumentParser()
    ap.add_argument("--name", default=os.environ.get("RAVNEST_NODE", default))
    ap.add_argument("--base-dir", default="node_data")
    args, _ = ap.parse_known_args()
    return args.name, args.base_dir
