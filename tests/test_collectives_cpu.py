"""Unit tier for the DP collectives: bucketed average_tensors /
average_optimizer_state vs an exact torch.mean reference (SURVEY.md
section 4 "ring-reduce numerical tests vs torch.mean"), 2 ranks, gloo.
"""
import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from ravnest_amd.comm.collectives import (average_tensors,
                                          average_optimizer_state,
                                          average_parameters)


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


def _master_worker(rank, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=2)
    g = torch.Generator().manual_seed(7 + rank)
    model = torch.nn.Linear(4, 4)
    with torch.no_grad():
        for p in model.parameters():
            p.copy_(torch.randn(p.shape, generator=g))
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    # fused-optimizer-style fp32 masters: the optimizer READS these on
    # every step, so averaging must cover them or DP sync is a no-op
    masters = {}
    for p in model.parameters():
        opt.state[p]["master"] = p.detach().float().clone() + rank
        masters[p] = opt.state[p]["master"]
    orig_masters = [m.clone() for m in masters.values()]
    average_parameters(model, dist.group.WORLD, optimizer=opt)
    torch.save({"orig_m": orig_masters,
                "avg_m": [m.clone() for m in masters.values()]},
               os.path.join(out_dir, f"master_{rank}.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_average_parameters_covers_fp32_masters(tmp_path):
    """Regression (round-1 advisor, collectives.py): with bf16 params +
    fp32 masters, averaging only the params is silently discarded at the
    next optimizer step. average_parameters must average masters too."""
    port = 29860 + (os.getpid() % 30)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_master_worker,
                         args=(r, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    for p in procs:
        assert p.exitcode == 0, p.exitcode
    res = {r: torch.load(tmp_path / f"master_{r}.pt") for r in range(2)}
    for m0, m1, o0, o1 in zip(res[0]["avg_m"], res[1]["avg_m"],
                              res[0]["orig_m"], res[1]["orig_m"]):
        ref = (o0 + o1) / 2
        assert torch.allclose(m0, ref, atol=1e-6), (m0 - ref).abs().max()
        assert torch.allclose(m1, ref, atol=1e-6)


def _async_worker(rank, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=2)
    from ravnest_amd.comm.collectives import AsyncReducer
    g = torch.Generator().manual_seed(40 + rank)
    ts = [torch.randn(16, 16, generator=g),
          torch.randn(257, generator=g)]
    orig = [t.clone() for t in ts]
    red = AsyncReducer(dist.group.WORLD, torch.device("cpu"))
    red.launch(ts)
    assert red.pending
    # tensors untouched until join (overlap window)
    assert red.join_into()
    assert not red.pending
    torch.save({"orig": orig, "avg": [t.clone() for t in ts]},
               os.path.join(out_dir, f"ar_{rank}.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_async_reducer_exact_mean(tmp_path):
    """Overlapped DP averaging: launch -> (overlap window) -> join must
    install the exact mean (same numerics as the synchronous path)."""
    port = 29820 + (os.getpid() % 30)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_async_worker, args=(r, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    for p in procs:
        assert p.exitcode == 0, p.exitcode
    res = {r: torch.load(tmp_path / f"ar_{r}.pt") for r in range(2)}
    for a0, a1, o0, o1 in zip(res[0]["avg"], res[1]["avg"],
                              res[0]["orig"], res[1]["orig"]):
        ref = (o0 + o1) / 2
        assert torch.allclose(a0, ref, atol=1e-6)
        assert torch.allclose(a1, ref, atol=1e-6)
