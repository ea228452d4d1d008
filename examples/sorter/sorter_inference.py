"""Offline pipelined inference: load the trained stage submodels
sequentially in one process and greedy-decode the sorted sequence
(parity: reference examples/sorter/sorter_inference.py:5-47)."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch  # noqa: E402

from examples.sorter.dataset import SortDataset  # noqa: E402


def load_stages(base_dir="node_data", cluster=0):
    stages = []
    s = 0
    while True:
        p = os.path.join(base_dir, f"cluster_{cluster}", f"stage_{s}")
        if not os.path.isdir(p):
            break
        gm = torch.load(os.path.join(p, "submod.pt"), weights_only=False)
        sd_path = os.path.join(p, "trained_state_dict.pt")
        if os.path.exists(sd_path):
            gm.load_state_dict(torch.load(sd_path, weights_only=True))
        gm.eval()
        stages.append(gm)
        s += 1
    return stages


def pipeline_forward(stages, x):
    out = x
    for st in stages:
        out = st(out) if not isinstance(out, tuple) else st(*out)
    return out[0] if isinstance(out, tuple) else out


if __name__ == "__main__":
    stages = load_stages()
    ds = SortDataset("test")
    length = ds.length
    n_ok = 0
    n = 50
    with torch.no_grad():
        for i in range(n):
            inp = ds.data[i]
            idx = inp.clone().unsqueeze(0)
            for _ in range(length):
                logits = pipeline_forward(stages, idx[:, -ds.block_size:])
                idx = torch.cat(
                    [idx, logits[:, -1, :].argmax(-1, keepdim=True)], 1)
            pred = idx[0, length:]
            n_ok += int(torch.equal(pred, torch.sort(inp)[0]))
    print(f"sorted correctly: {n_ok}/{n}")
