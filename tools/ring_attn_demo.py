"""Long-context ring attention demo: 2 ranks sharing one MI355X, each
holding half of an 8192-token sequence (BERT-base head geometry), fused
flash kernels per chunk, exactness vs a single-process full-sequence
run. Launch: torchrun --nproc-per-node 2 tools/ring_attn_demo.py
"""
import math
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402


def main():
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29555")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from ravnest_amd.parallel import ring_attention
    dev = torch.device("cuda", 0)
    B, H, Sg, D = 4, 12, 8192, 64
    Sl = Sg // world
    torch.manual_seed(3)
    q = torch.randn(B, H, Sg, D).to(dev, torch.bfloat16)
    k = torch.randn(B, H, Sg, D).to(dev, torch.bfloat16)
    v = torch.randn(B, H, Sg, D).to(dev, torch.bfloat16)
    sl = slice(rank * Sl, (rank + 1) * Sl)
    ql, kl, vl = (t[:, :, sl].contiguous() for t in (q, k, v))
    # timed fwd (5 iters after 2 warmup)
    for _ in range(2):
        o = ring_attention(ql, kl, vl, causal=True)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(5):
        o = ring_attention(ql, kl, vl, causal=True)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 5 * 1000
    # exactness vs single-process full attention (rank 0 only)
    if rank == 0:
        from ravnest_amd.ops import get_ext
        ext = get_ext(True)
        o_full, _ = ext.attn_fwd(q, k, v, torch.Tensor(), True,
                                 1.0 / math.sqrt(D))
        err = (o.float() - o_full[:, :, sl].float()).abs().max().item()
        flops = 2 * 2 * B * H * (Sg * Sg / 2) * D / world  # causal, per rank
        print(f"ring_attention S_global={Sg} world={world} "
              f"B={B} H={H} D={D}: {ms:.2f} ms/fwd per rank "
              f"({flops / ms / 1e9:.0f} TF/s incl. ring shifts), "
              f"max |err| vs full-sequence kernel = {err:.2e}",
              flush=True)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
