"""Sequence-parallel GPT vs the unsharded model: logits of every shard
and (all-reduced) parameter gradients must match a single-process run
of the SAME weights (ravnest_amd/parallel/sp.py)."""
import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from ravnest_amd import set_seed
from ravnest_amd.models.gpt import GPT, GPTConfig


def _build(seed=21):
    set_seed(seed)
    # head_dim 64 (the kernels' native size on GPU; CPU uses math path),
    # no dropout so SP and reference runs are deterministic-identical
    cfg = GPTConfig(vocab_size=64, block_size=32, n_layer=2, n_head=2,
                    n_embd=128, dropout=0.0)
    return GPT(cfg), cfg


def _worker(rank, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=2)
    from ravnest_amd.parallel import (allreduce_gradients,
                                      sequence_parallelize)
    model, cfg = _build()
    sequence_parallelize(model)
    torch.manual_seed(99)
    B, Sg = 2, cfg.block_size
    Sl = Sg // 2
    idx = torch.randint(0, cfg.vocab_size, (B, Sg))
    logits = model(idx[:, rank * Sl:(rank + 1) * Sl])
    targets = torch.roll(idx, -1, 1)[:, rank * Sl:(rank + 1) * Sl]
    # per-token-summed CE so shard losses add up to the full-run loss
    loss = torch.nn.functional.cross_entropy(
        logits.reshape(-1, cfg.vocab_size), targets.reshape(-1),
        reduction="sum")
    loss.backward()
    allreduce_gradients(model)
    g_tok = model.embeddings.tok_emb.weight.grad.clone()
    g_qkv = model.blocks[0].qkv.weight.grad.clone()
    g_head = model.lm_head.weight.grad.clone()
    torch.save({"logits": logits.detach(), "loss": loss.detach(),
                "g_tok": g_tok, "g_qkv": g_qkv, "g_head": g_head},
               os.path.join(out_dir, f"sp_{rank}.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_sequence_parallel_gpt(tmp_path):
    port = 29780 + (os.getpid() % 30)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
    for p in procs:
        assert p.exitcode == 0, p.exitcode

    # unsharded reference with the SAME weights and data
    model, cfg = _build()
    torch.manual_seed(99)
    B, Sg = 2, cfg.block_size
    Sl = Sg // 2
    idx = torch.randint(0, cfg.vocab_size, (B, Sg))
    logits = model(idx)
    targets = torch.roll(idx, -1, 1)
    loss = torch.nn.functional.cross_entropy(
        logits.reshape(-1, cfg.vocab_size), targets.reshape(-1),
        reduction="sum")
    loss.backward()

    res = {r: torch.load(tmp_path / f"sp_{r}.pt") for r in range(2)}
    # logits shards
    for r in range(2):
        sl = slice(r * Sl, (r + 1) * Sl)
        err = (res[r]["logits"] - logits[:, sl].detach()).abs().max()
        assert err < 1e-4, f"rank {r} logits err {err}"
    # shard losses sum to the full loss
    tot = res[0]["loss"] + res[1]["loss"]
    assert torch.allclose(tot, loss.detach(), rtol=1e-5), (tot, loss)
    # all-reduced parameter grads == full-model grads (both ranks hold
    # the same reduced copy)
    for name, ref in [("g_tok", model.embeddings.tok_emb.weight.grad),
                      ("g_qkv", model.blocks[0].qkv.weight.grad),
                      ("g_head", model.lm_head.weight.grad)]:
        for r in range(2):
            err = (res[r][name] - ref).abs().max()
            assert err < 1e-4, f"rank {r} {name} err {err}"


def _gpu_worker(rank, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=2)
    from ravnest_amd.parallel import sequence_parallelize
    model, cfg = _build()
    model = model.to("cuda", torch.bfloat16)
    sequence_parallelize(model)
    torch.manual_seed(99)
    B, Sg = 2, cfg.block_size
    Sl = Sg // 2
    idx = torch.randint(0, cfg.vocab_size, (B, Sg)).cuda()
    logits = model(idx[:, rank * Sl:(rank + 1) * Sl])
    torch.save({"logits": logits.detach().float().cpu()},
               os.path.join(out_dir, f"spg_{rank}.pt"))
    dist.barrier()
    dist.destroy_process_group()


import pytest  # noqa: E402


@pytest.mark.gpu
def test_sequence_parallel_gpt_gpu(tmp_path):
    """Whole-model SP forward on GPU (bf16, ring kernel chunks, eager
    embedding offsets) vs the single-process bf16 model."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    port = 29750 + (os.getpid() % 30)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_gpu_worker, args=(r, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
    for p in procs:
        assert p.exitcode == 0, p.exitcode
    model, cfg = _build()
    model = model.to("cuda", torch.bfloat16)
    torch.manual_seed(99)
    B, Sg = 2, cfg.block_size
    Sl = Sg // 2
    idx = torch.randint(0, cfg.vocab_size, (B, Sg)).cuda()
    ref = model(idx).float().cpu()
    scale = ref.abs().max().item()
    for r in range(2):
        got = torch.load(tmp_path / f"spg_{r}.pt")["logits"]
        sl = slice(r * Sl, (r + 1) * Sl)
        err = (got - ref[:, sl]).abs().max().item()
        assert err < 0.05 * scale + 0.05, f"rank {r} err {err} scale {scale}"
