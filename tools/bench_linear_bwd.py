"""A/B the linear fwd+bwd variants on the BERT-base projection shapes.

Motivation (profiles/r02_bert_base_bs128_default.txt): the default
nn.Linear path spends 2.2% of the step in at::native::reduce_kernel and
1.1% in __amd_rocclr_copyBuffer, ~37 calls/step each — the bias-grad
backward. ops.Linear replaces the reduce with the hand colsum kernel but
measured WORSE overall in round 1 (autograd's addmm/matmul backward hits
better hipBLASLt wgrad selections). This bench isolates each piece.

Run: python tools/bench_linear_bwd.py
"""
import os
import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
from ravnest_amd.ops import get_ext  # noqa: E402


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(iters):
        fn()
    t1.record()
    torch.cuda.synchronize()
    return t0.elapsed_time(t1) / iters


def main():
    ext = get_ext(required=True)
    dev = "cuda"
    torch.manual_seed(0)
    B, S = 128, 512
    M = B * S
    shapes = [("qkv", 768, 2304), ("attn_out", 768, 768),
              ("mlp_out", 3072, 768), ("head_dense", 768, 768)]
    for name, K, N in shapes:
        x3 = torch.randn(B, S, K, device=dev, dtype=torch.bfloat16,
                         requires_grad=True)
        w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02
        bias = torch.zeros(N, device=dev, dtype=torch.bfloat16)
        dy3 = torch.randn(B, S, N, device=dev, dtype=torch.bfloat16)
        w.requires_grad_(True)
        bias.requires_grad_(True)

        def v0():  # nn.Linear-equivalent autograd
            for t in (x3, w, bias):
                t.grad = None
            y = F.linear(x3, w, bias)
            y.backward(dy3)

        x2 = x3.detach().reshape(M, K).contiguous()
        dy2 = dy3.reshape(M, N).contiguous()

        def v1():  # explicit: dx = dy@w, dw = dy^T@x, db = colsum
            dx = (dy2 @ w.detach()).view_as(x3)
            dw = dy2.t() @ x2
            db = ext.colsum_bf16(dy2)
            return dx, dw, db

        def v2():  # autograd-mimic wgrad: mm on transposed views
            dx = torch.matmul(dy3, w.detach())
            dw = torch.mm(x2.t(), dy2).t().contiguous()
            db = ext.colsum_bf16(dy2)
            return dx, dw, db

        def v3():  # fwd-only (for reference)
            return F.linear(x3.detach(), w.detach(), bias.detach())

        def v4():  # torch bias grad alone
            return dy3.sum((0, 1))

        def v5():  # colsum alone
            return ext.colsum_bf16(dy2)

        r = {"v0_autograd_fb": timeit(v0), "v1_explicit": timeit(v1),
             "v2_mimic": timeit(v2), "v3_fwd": timeit(v3),
             "v4_torch_db": timeit(v4), "v5_colsum_db": timeit(v5)}
        line = " ".join(f"{k}={v:.3f}ms" for k, v in r.items())
        print(f"{name:10s} K={K} N={N}: {line}", flush=True)
        # bwd-only cost of the autograd path:
        print(f"{name:10s} v0_bwd_est={r['v0_autograd_fb']-r['v3_fwd']:.3f}ms"
              f"  v1={r['v1_explicit']:.3f}  v2={r['v2_mimic']:.3f}",
              flush=True)

    # where do the copyBuffers come from? profile one nn.Linear step
    x3 = torch.randn(B, S, 768, device=dev, dtype=torch.bfloat16,
                     requires_grad=True)
    w = torch.randn(2304, 768, device=dev, dtype=torch.bfloat16,
                    requires_grad=True)
    bias = torch.zeros(2304, device=dev, dtype=torch.bfloat16,
                       requires_grad=True)
    dy3 = torch.randn(B, S, 2304, device=dev, dtype=torch.bfloat16)
    y = F.linear(x3, w, bias)
    y.backward(dy3)  # warm
    from torch.profiler import profile, ProfilerActivity
    with profile(activities=[ProfilerActivity.CUDA]) as prof:
        for _ in range(3):
            x3.grad = w.grad = bias.grad = None
            y = F.linear(x3, w, bias)
            y.backward(dy3)
    print(prof.key_averages().table(sort_by="cuda_time_total", row_limit=18))


if __name__ == "__main__":
    main()
