#!/usr/bin/env python3
"""hipBLASLt (torch.matmul) throughput on the BERT-base step's GEMM
shapes, bf16. Establishes whether a hand-written CDNA4 GEMM (guide's
256^2 8-phase template: ~1.33-1.47 PF on random data) would beat the
library on these shapes."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch  # noqa: E402


def timeit(fn, iters=30, warmup=8):
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
    return s.elapsed_time(e) / iters


def main():
    dev = torch.device("cuda", 0)
    T = 32 * 512  # tokens at micro-batch 32
    shapes = [
        ("qkv fwd    ", T, 2304, 768, "nt"),
        ("attn_out   ", T, 768, 768, "nt"),
        ("mlp_in fwd ", T, 3072, 768, "nt"),
        ("mlp_out fwd", T, 768, 3072, "nt"),
        ("head fwd   ", T, 30522, 768, "nt"),
        ("qkv dgrad  ", T, 768, 2304, "nn"),
        ("mlp dgrad  ", T, 3072, 768, "nn"),
        ("qkv wgrad  ", 2304, 768, T, "tn"),
        ("mlp wgrad  ", 3072, 768, T, "tn"),
        ("head wgrad ", 30522, 768, T, "tn"),
    ]
    for name, M, N, K, layout in shapes:
        if layout == "nt":  # x @ W.T  (linear fwd)
            a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
            b = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
            fn = lambda: a @ b.t()  # noqa: E731
        elif layout == "nn":  # dy @ W (dgrad)
            a = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
            b = torch.randn(K, N, device=dev, dtype=torch.bfloat16)
            fn = lambda: a @ b  # noqa: E731
        else:  # tn: dy.T @ x (wgrad)
            a = torch.randn(K, M, device=dev, dtype=torch.bfloat16)
            b = torch.randn(K, N, device=dev, dtype=torch.bfloat16)
            fn = lambda: a.t() @ b  # noqa: E731
        ms = timeit(fn)
        tf = 2 * M * N * K / ms / 1e9
        print(f"{name} M={M:6d} N={N:6d} K={K:6d} {layout}: "
              f"{ms:7.3f} ms  {tf:7.1f} TF/s")


if __name__ == "__main__":
    main()
