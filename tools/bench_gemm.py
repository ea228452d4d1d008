"""A/B the hand-written 8-phase bf16 GEMM against hipBLASLt (torch
matmul) on the BERT-base bench shapes. Run on a GPU box:

    python tools/bench_gemm.py [--iters 20]
"""
import argparse
import math
import sys
import os

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402


def time_fn(fn, iters, warmup=5):
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
    ap.add_argument("--iters", type=int, default=20)
    ap.add_argument("--shapes", type=str, default="")
    args = ap.parse_args()
    from ravnest_amd.ops import get_ext
    ext = get_ext(True)
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)

    shapes = [
        # (M, K, N, label) — BERT-base bs128 seq512 forward projections
        (65536, 768, 2304, "qkv"),
        (65536, 768, 768, "attn_out"),
        (65536, 768, 3072, "mlp_in"),
        (65536, 3072, 768, "mlp_out"),
        (65536, 768, 30522, "mlm_head"),
        # dgrad shapes (K = prev N)
        (65536, 2304, 768, "qkv_dgrad"),
        (65536, 3072, 768, "mlp_in_dgrad"),
        (8192, 768, 768, "small_M"),
    ]
    if args.shapes:
        shapes = [tuple(int(x) for x in s.split("x")) + (s,)
                  for s in args.shapes.split(",")]

    for (M, K, N, label) in shapes:
        a = (torch.randn(M, K, device=dev) / math.sqrt(K)).to(torch.bfloat16)
        b = torch.randn(N, K, device=dev).to(torch.bfloat16)
        bt = b.t().contiguous().t()  # col-major view for addmm path parity
        flops = 2.0 * M * K * N

        t_ours = time_fn(lambda: ext.gemm_nt_bf16(a, b, None, 0), args.iters)
        t_blas = time_fn(lambda: torch.matmul(a, b.t()), args.iters)
        # correctness spot check on a slice
        (c,) = ext.gemm_nt_bf16(a, b, None, 0)
        ref = torch.matmul(a[:256].float(), b.float().t())
        err = (c[:256].float() - ref).abs().max().item() / \
            (ref.abs().max().item() + 1e-9)
        print(f"{label:14s} M{M} K{K} N{N}: ours {t_ours:7.3f} ms "
              f"({flops/t_ours/1e9:7.1f} TF) | blaslt {t_blas:7.3f} ms "
              f"({flops/t_blas/1e9:7.1f} TF) | ratio "
              f"{t_blas/t_ours:5.2f}x | relerr {err:.3e}", flush=True)
        del a, b, bt


def bench_wgrad(iters=10):
    """wgrad dW[N,K] = dy[M,N]^T @ x[M,K]: library vs transpose+hand-NT
    (reduction M=65536 becomes the hand kernel's deep-K dimension)."""
    from ravnest_amd.ops import get_ext
    ext = get_ext(True)
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    M = 65536
    for (N, K, label) in [(2304, 768, "qkv_w"), (768, 768, "attn_out_w"),
                          (3072, 768, "mlp_in_w"), (768, 3072, "mlp_out_w"),
                          (30522, 768, "mlm_head_w")]:
        dy = (torch.randn(M, N, device=dev) / 64).to(torch.bfloat16)
        x = (torch.randn(M, K, device=dev) / 64).to(torch.bfloat16)
        flops = 2.0 * M * K * N

        def lib():
            return dy.t() @ x

        def hand():
            return ext.gemm_wgrad_bf16(dy, x).to(torch.bfloat16)

        t_lib = time_fn(lib, iters)
        t_hand = time_fn(hand, iters)
        c = hand().float()
        cref = (dy.t().contiguous().float() @ x.float())
        err = (c - cref).abs().max().item() / (cref.abs().max().item() + 1e-9)
        print(f"{label:12s} N{N} K{K} M{M}: hand {t_hand:7.3f} ms "
              f"({flops/t_hand/1e9:7.1f} TF) | lib {t_lib:7.3f} ms "
              f"({flops/t_lib/1e9:7.1f} TF) | ratio {t_lib/t_hand:5.2f}x "
              f"| relerr {err:.3e}", flush=True)
        del dy, x





def bench_mx(iters=20):
    from ravnest_amd.ops import get_ext
    ext = get_ext(True)
    dev = torch.device("cuda", 0)
    torch.manual_seed(0)
    for (M, K, N, label) in [(32768, 768, 2304, "gpt_qkv"),
                             (32768, 768, 3072, "gpt_mlp_in"),
                             (32768, 3072, 768, "gpt_mlp_out"),
                             (4096, 4096, 4096, "square4k"),
                             (8192, 8192, 8192, "square8k")]:
        x = (torch.randn(M, K, device=dev) / 8).to(torch.bfloat16)
        w = (torch.randn(N, K, device=dev) / 8).to(torch.bfloat16)
        xq, xs = ext.mx_quant(x)
        wq, ws = ext.mx_quant(w)
        flops = 2.0 * M * K * N
        t1 = time_fn(lambda: ext.mx_gemm(xq, xs, wq, ws), iters)
        t2 = time_fn(lambda: ext.mx_gemm2(xq, xs, wq, ws), iters)
        tb = time_fn(lambda: torch.matmul(x, w.t()), iters)
        print(f"{label:11s} M{M} K{K} N{N}: mx2 {t2:7.3f} ms "
              f"({flops/t2/1e9:7.1f} TF) | mx1 {t1:7.3f} "
              f"({flops/t1/1e9:7.1f}) | bf16-lib {tb:7.3f} "
              f"({flops/tb/1e9:7.1f})", flush=True)
        del x, w, xq, wq


if __name__ == "__main__":
    if "--wgrad" in sys.argv:
        bench_wgrad()
    elif "--mx" in sys.argv:
        bench_mx()
    else:
        main()
