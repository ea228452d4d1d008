"""Unit tier for the DP collectives: bucketed average_tensors /
average_optimizer_state vs an exact torch.mean reference (SURVEY.md
section 4 "ring-reduce numerical tests vs torch.mean"), 2 ranks, gloo.
"""
import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from ravnest_amd.comm.collectives import (average_tensors,
                                          average_optimizer_state)


def _worker(rank, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=2)
    g = torch.Generator().manual_seed(100 + rank)
    # mixed dtypes + a tensor larger than the bucket to force a flush
    tensors = [torch.randn(3, 5, generator=g),
               torch.randn(1000, generator=g).double(),
               torch.randn(64, 64, generator=g)]
    originals = [t.clone() for t in tensors]
    average_tensors(tensors, dist.group.WORLD, bucket_bytes=4096)

    # optimizer-state averaging: Adam moments on a dummy param
    p = torch.nn.Parameter(torch.randn(8, 8, generator=g))
    opt = torch.optim.Adam([p], lr=1e-3)
    p.grad = torch.randn(8, 8, generator=g)
    opt.step()
    m0 = opt.state[p]["exp_avg"].clone()
    average_optimizer_state(opt, dist.group.WORLD)
    torch.save({"orig": originals, "avg": tensors, "m0": m0,
                "mavg": opt.state[p]["exp_avg"].clone()},
               os.path.join(out_dir, f"coll_{rank}.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_average_tensors_matches_mean(tmp_path):
    port = 29960 + (os.getpid() % 30)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    for p in procs:
        assert p.exitcode == 0, p.exitcode
    res = {r: torch.load(tmp_path / f"coll_{r}.pt") for r in range(2)}
    for a0, a1, o0, o1 in zip(res[0]["avg"], res[1]["avg"],
                              res[0]["orig"], res[1]["orig"]):
        ref = (o0 + o1) / 2
        assert torch.allclose(a0, ref, atol=1e-6), (a0 - ref).abs().max()
        assert torch.allclose(a1, ref, atol=1e-6)
    mref = (res[0]["m0"] + res[1]["m0"]) / 2
    assert torch.allclose(res[0]["mavg"], mref, atol=1e-6)
    assert torch.allclose(res[1]["mavg"], mref, atol=1e-6)
