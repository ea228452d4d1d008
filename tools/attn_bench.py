#!/usr/bin/env python3
"""Attention kernel micro-bench: fwd/bwd wall time + effective TF/s for
both layouts (separate (B,H,S,D) and packed (B,S,3,H,D)).

Usage (GPU box): python tools/attn_bench.py [--B 32] [--S 512] [--H 12]
"""
import argparse
import math
import sys
import os

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters  # ms


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--B", type=int, default=32)
    ap.add_argument("--S", type=int, default=512)
    ap.add_argument("--H", type=int, default=12)
    ap.add_argument("--D", type=int, default=64)
    args = ap.parse_args()
    from ravnest_amd.ops import get_ext
    ext = get_ext(True)
    dev = torch.device("cuda", 0)
    B, S, H, D = args.B, args.S, args.H, args.D
    scale = 1.0 / math.sqrt(D)
    torch.manual_seed(0)

    # FLOPs: fwd 2 matmuls, bwd 7 matmul passes (S,dP twice + dV,dK,dQ)
    mm = 2 * B * H * S * S * D  # one S x S x D matmul pass (x2 flops)
    fwd_fl, bwd_fl = 2 * mm, 7 * mm

    q, k, v = (torch.randn(B, H, S, D, device=dev, dtype=torch.bfloat16)
               for _ in range(3))
    empty = torch.Tensor()
    o, lse = ext.attn_fwd(q, k, v, empty, False, scale)
    do = torch.randn_like(o)
    t_f = timeit(lambda: ext.attn_fwd(q, k, v, empty, False, scale))
    t_b = timeit(lambda: ext.attn_bwd(q, k, v, o, do, lse, empty, False,
                                      scale))
    print(f"separate: fwd {t_f:7.3f} ms ({fwd_fl/t_f/1e9:7.1f} TF/s)  "
          f"bwd {t_b:7.3f} ms ({bwd_fl/t_b/1e9:7.1f} TF/s)")

    qkv = torch.randn(B, S, 3, H, D, device=dev, dtype=torch.bfloat16)
    o2, lse2 = ext.attn_fwd_qkv(qkv, empty, False, scale)
    do2 = torch.randn_like(o2)
    t_f2 = timeit(lambda: ext.attn_fwd_qkv(qkv, empty, False, scale))
    t_b2 = timeit(lambda: ext.attn_bwd_qkv(qkv, o2, do2, lse2, empty, False,
                                           scale))
    print(f"packed:   fwd {t_f2:7.3f} ms ({fwd_fl/t_f2/1e9:7.1f} TF/s)  "
          f"bwd {t_b2:7.3f} ms ({bwd_fl/t_b2/1e9:7.1f} TF/s)")


if __name__ == "__main__":
    main()
