"""fp8 wire compression: unit round-trip of the pack/unpack helpers and a
2-process gloo Channel exchange with RAVNEST_WIRE_FP8=1 (extends the
reference's lossy fp16 wire compression, ravnest/utils.py:184-194, to
the CDNA4-native scaled-e4m3 format — comm/p2p.py)."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from ravnest_amd.comm.p2p import (Channel, Message, _f8_pack, _f8_unpack,
                                  _F8_DTYPE, _MAX_DIMS)
from ravnest_amd.strings import ActionTypes


@pytest.mark.skipif(_F8_DTYPE is None, reason="no float8 in this torch")
def test_f8_pack_roundtrip():
    torch.manual_seed(0)
    for dtype in (torch.float32, torch.bfloat16, torch.float16):
        t = torch.randn(17, 33, dtype=dtype) * 123.0
        q, scale, orig = _f8_pack(t)
        assert q.dtype == torch.uint8 and orig == dtype
        r = _f8_unpack(q, scale, orig)
        assert r.dtype == dtype and r.shape == t.shape
        amax = t.float().abs().max()
        err = (r.float() - t.float()).abs().max()
        assert err <= amax * 0.04, f"{dtype}: err {err} amax {amax}"
    # zero tensor: scale falls back to 1, exact zeros back
    z = torch.zeros(4, 4)
    q, scale, orig = _f8_pack(z)
    assert scale == 1.0
    assert (_f8_unpack(q, scale, orig) == 0).all()
    # ineligible shapes/dtypes fall back to the plain wire
    assert _f8_pack(torch.zeros([2] * _MAX_DIMS)) is None
    assert _f8_pack(torch.zeros(3, dtype=torch.int64)) is None


def _worker(rank, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RAVNEST_WIRE_FP8"] = "1"
    dist.init_process_group("gloo", rank=rank, world_size=2)
    got = {}

    def deliver(ch, msg):
        got["msg"] = msg

    ch = Channel(src=0, dst=1, kind="fwd", group=dist.group.WORLD,
                 device=torch.device("cpu"), my_rank=rank, deliver=deliver)
    ch.start()
    torch.manual_seed(7)
    acts = torch.randn(2, 8, 16, dtype=torch.bfloat16) * 50
    grads = torch.randn(4, 4) * 3
    ids = torch.arange(6)  # ints must pass through exactly
    if rank == 0:
        ch.send(Message(action=ActionTypes.FORWARD, fpid=3,
                        tensors=[(1, acts), (2, grads), (3, ids)], extra=9))
        ch.close()
        ch.join(timeout=60)
    else:
        ch.join(timeout=60)
        msg = got["msg"]
        assert msg.action == ActionTypes.FORWARD
        assert msg.fpid == 3 and msg.extra == 9
        (g1, r_acts), (g2, r_grads), (g3, r_ids) = msg.tensors
        assert (g1, g2, g3) == (1, 2, 3)
        assert r_acts.dtype == torch.bfloat16
        assert r_grads.dtype == torch.float32
        torch.save({"acts": r_acts, "grads": r_grads, "ids": r_ids},
                   os.path.join(out_dir, "rx.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_channel_fp8_wire(tmp_path):
    port = 29860 + (os.getpid() % 30)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    for p in procs:
        assert p.exitcode == 0, p.exitcode
    rx = torch.load(tmp_path / "rx.pt")
    torch.manual_seed(7)
    acts = torch.randn(2, 8, 16, dtype=torch.bfloat16) * 50
    grads = torch.randn(4, 4) * 3
    a_err = (rx["acts"].float() - acts.float()).abs().max()
    assert a_err <= acts.float().abs().max() * 0.04, a_err
    g_err = (rx["grads"] - grads).abs().max()
    assert g_err <= grads.abs().max() * 0.04, g_err
    assert (rx["ids"] == torch.arange(6)).all()


def _gpu_worker(rank, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RAVNEST_WIRE_FP8"] = "1"
    dist.init_process_group("gloo", rank=rank, world_size=2)
    got = {}

    def deliver(ch, msg):
        got["msg"] = msg

    dev = torch.device("cuda", 0)
    ch = Channel(src=0, dst=1, kind="fwd", group=dist.group.WORLD,
                 device=dev, my_rank=rank, deliver=deliver)
    ch.start()
    torch.manual_seed(9)
    acts = (torch.randn(4, 16, 64) * 30).to(dev, torch.bfloat16)
    if rank == 0:
        ch.send(Message(action=ActionTypes.FORWARD, fpid=1,
                        tensors=[(7, acts)]))
        ch.close()
        ch.join(timeout=60)
    else:
        ch.join(timeout=60)
        (gid, rx) = got["msg"].tensors[0]
        assert gid == 7 and rx.is_cuda and rx.dtype == torch.bfloat16
        err = (rx.float() - acts.float()).abs().max()
        amax = acts.float().abs().max()
        assert err <= amax * 0.04, f"{err} vs {amax}"
        torch.save({"ok": True}, os.path.join(out_dir, "f8g.pt"))
    dist.barrier()
    dist.destroy_process_group()


import pytest  # noqa: E402


@pytest.mark.gpu
def test_channel_fp8_wire_gpu(tmp_path):
    """fp8 wire with CUDA payloads: quantize on device, host-staged
    gloo transport (wire_cpu), reconstruct on the receiver's GPU."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import torch.multiprocessing as mp2
    port = 29690 + (os.getpid() % 30)
    ctx = mp2.get_context("spawn")
    procs = [ctx.Process(target=_gpu_worker, args=(r, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
    for p in procs:
        assert p.exitcode == 0, p.exitcode
    assert (tmp_path / "f8g.pt").exists()
