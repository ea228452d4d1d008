"""Offline planning driver (parity: reference cluster_formation.py):
clusterize the chosen workload into node_data/ artifacts. The CNN
walkthrough is active; others mirror the reference's commented blocks —
pass --model to pick one."""
import argparse

import torch

from ravnest_amd import clusterize, set_seed
from ravnest_amd.models import (CNN, BertConfig, BertForMLM, GPT, GPTConfig,
                                Inception3, resnet50)

set_seed(42)

if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="cnn",
                    choices=["cnn", "resnet50", "inception_v3", "sorter",
                             "bert"])
    ap.add_argument("--node-config", default="node_data/node_configs.json")
    ap.add_argument("--base-dir", default="node_data")
    ap.add_argument("--max-clusters", type=int, default=5)
    ap.add_argument("--fp8", action="store_true",
                    help="sorter: MX fp8 block projections")
    args = ap.parse_args()

    if args.model == "cnn":
        model = CNN()
        ex = (torch.randn(2, 1, 8, 8),)
    elif args.model == "resnet50":
        model = resnet50()
        ex = (torch.randn(2, 3, 64, 64),)
    elif args.model == "inception_v3":
        model = Inception3()
        ex = (torch.randn(2, 3, 32, 32),)
    elif args.model == "sorter":
        from examples.sorter.dataset import SortDataset
        ds = SortDataset("train")
        # --fp8 plans the sorter with MX fp8 block projections
        # (head_dim-64 scale so the attention kernel + MXLinear apply)
        cfg = (GPTConfig.nano64(vocab_size=ds.vocab_size,
                                block_size=ds.block_size)
               if args.fp8 else
               GPTConfig.nano(vocab_size=ds.vocab_size,
                              block_size=ds.block_size))
        cfg.fp8 = args.fp8
        model = GPT(cfg)
        ex = (ds[0][0].unsqueeze(0),)
    else:  # bert
        cfg = BertConfig.base()
        model = BertForMLM(cfg)
        ids = torch.randint(0, cfg.vocab_size, (2, 512))
        ex = (ids, torch.ones_like(ids))

    meta = clusterize(model, ex, node_config_path=args.node_config,
                      max_clusters=args.max_clusters, base_dir=args.base_dir)
    print("planned:", meta)
