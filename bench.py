#!/usr/bin/env python3
"""Flagship benchmark: BERT-base MLM pretraining step on 1..8 MI355X.

Driver contract (BASELINE.json): metric = samples/sec (whole node) for the
reference's BERT-base pretraining workload (examples/bert/provider.py
config family), async DP via the ravnest_amd Node (one fused
root+leaf replica per GPU, periodic RCCL parameter averaging — the
reference's cross-cluster ring-averaging semantics). Weak scaling:
per-GPU micro-batch fixed as N grows.

Launch:
  python bench.py --gpus 1 --steps 20 --warmup 5
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Rank 0 prints ONE JSON line with the whole-job aggregate.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

# hipBLASLt algorithm selection pre-tuned on MI355X (TunableOp, +3% on the
# BERT-base step). Must be set before the first GEMM; tuning stays off.
_TUNED = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "tools", "tunableop_mi355x.csv")
if os.path.exists(_TUNED) and "PYTORCH_TUNABLEOP_ENABLED" not in os.environ:
    os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
    os.environ["PYTORCH_TUNABLEOP_TUNING"] = "0"
    os.environ["PYTORCH_TUNABLEOP_FILENAME"] = _TUNED

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from ravnest_amd import Node, set_seed  # noqa: E402
from ravnest_amd.comm import CommBackend  # noqa: E402
from ravnest_amd.ops import CrossEntropyLoss, FusedAdam  # noqa: E402
from ravnest_amd.models import BertConfig, BertForMLM, GPTConfig, GPT  # noqa: E402


def build_model(name: str, seq: int):
    if name == "bert-base":
        return BertForMLM(BertConfig.base(max_seq=seq)), "bert"
    if name == "bert-tiny":
        return BertForMLM(BertConfig.tiny(max_seq=seq)), "bert"
    if name == "gpt2-small":
        return GPT(GPTConfig.gpt2_small(block_size=seq)), "gpt"
    if name == "gpt-nano":
        return GPT(GPTConfig.nano64(block_size=seq)), "gpt"
    if name == "gpt2-small-fp8":
        # BASELINE.json config 5: block projections on the MX fp8 MFMA
        # path (ops.MXLinear -> csrc/mx_gemm2.hip)
        cfg = GPTConfig.gpt2_small(block_size=seq)
        cfg.fp8 = True
        return GPT(cfg), "gpt"
    if name == "resnet50":
        from ravnest_amd.models import resnet50
        m = resnet50(num_classes=200)
        m.cfg = type("C", (), {"vocab_size": 200})()  # criterion range
        return m, "resnet"
    raise ValueError(name)


class _ListLoader:
    def __init__(self, items):
        self.items = items

    def __iter__(self):
        return iter(self.items)


def run_pp(args, rank, world, device, on_gpu, family, model, batches,
           labels, share_gpu=False):
    """Async pipeline-parallel bench: ONE model split into `world` stages
    (1 cluster x N stages — the reference's headline topology). Rank 0
    plans with clusterize() and drives microbatch injection; every stage
    reports the barrier-bracketed timed window (MAX over ranks)."""
    import time as _time
    from ravnest_amd import clusterize
    from ravnest_amd.planner import NodeSpec

    base = os.path.abspath(f"bench_pp_plan_w{world}")
    ready = os.path.join(base, ".ready")
    if rank == 0:
        pool = [NodeSpec(name=f"n{i}", ram=8 * 2**30) for i in range(world)]
        plan_model = build_model(args.model, args.seq)[0]  # fp32 for fx
        ex = batches[0]
        ex_args = tuple(v.cpu() for v in ex.values())
        clusterize(plan_model, ex_args, node_pool=pool, max_clusters=1,
                   base_dir=base)
        with open(ready, "w") as f:
            f.write("ok")
    else:
        deadline = time.time() + 600
        me = os.path.join(base, "nodes", f"node_{rank}.json")
        while not (os.path.exists(ready) and os.path.exists(me)):
            if time.time() > deadline:
                raise TimeoutError("pp plan did not appear")
            time.sleep(0.5)

    transform = ((lambda m: m.to(torch.bfloat16))
                 if args.dtype == "bf16" and on_gpu else None)
    from ravnest_amd.ops import CrossEntropyLoss, FusedAdam
    node = Node(name=f"node_{rank}", base_dir=base,
                optimizer=FusedAdam if on_gpu else torch.optim.Adam,
                optimizer_params={"lr": 1e-4},
                device=device,
                backend="gloo" if share_gpu else None,
                criterion=CrossEntropyLoss(-100),
                labels=_ListLoader([labels[i % len(labels)] for i in
                                    range(args.warmup + args.steps +
                                          len(labels))]),
                update_frequency=1,
                model_transform=transform,
                loss_filename="bench_pp_losses.txt")
    node.start()
    import torch.distributed as dist

    def sync():
        if on_gpu:
            torch.cuda.synchronize(device)

    is_root = node.node_type.value == "root"
    dist.barrier()
    if is_root:
        for i in range(args.warmup):
            node.forward_compute(**batches[i % len(batches)])
        node.wait_for_backwards()
    sync()
    dist.barrier()
    sync()
    t0 = _time.perf_counter()
    if is_root:
        for i in range(args.steps):
            node.forward_compute(**batches[i % len(batches)])
        node.wait_for_backwards()
    sync()
    dist.barrier()
    elapsed = _time.perf_counter() - t0
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=device if on_gpu and not share_gpu else "cpu")
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    if rank == 0:
        samples_per_sec = args.micro_batch * args.steps / elapsed
        print(json.dumps({
            "metric": "samples/sec (whole node), "
                      f"{args.model} async pipeline",
            "value": round(samples_per_sec, 2),
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 2),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": args.dtype if on_gpu else "fp32",
            "data": "synthetic",
            "config": {"model": args.model,
                       "global_batch": args.micro_batch,
                       "seq_len": args.seq,
                       "parallelism": f"pp{world}",
                       "optimizer": "fused_adam" if on_gpu else "adam"},
        }), flush=True)
        node.stop_cluster()
    node.stop()
    sys.stdout.flush()
    os._exit(0)


def run_sp(args, rank, world, device, on_gpu, family, model, batches,
           labels, share_gpu=False):
    """Sequence-parallel bench: ONE GPT with the sequence sharded across
    the ranks — ring attention in every block, token-local everything
    else, parameter grads all-reduced over the SP group
    (ravnest_amd/parallel/sp.py). Strong scaling: the global
    (batch, seq) problem is fixed while ranks grow."""
    import time as _time
    import torch.distributed as dist
    if family != "gpt":
        raise SystemExit("--parallelism sp supports the GPT family")
    if args.seq % world:
        raise SystemExit("--seq must divide by the SP world size")
    backend = "gloo" if (share_gpu or not on_gpu) else "nccl"
    dist.init_process_group(backend, rank=rank, world_size=world)
    from ravnest_amd.parallel import (allreduce_gradients,
                                      sequence_parallelize)
    sequence_parallelize(model)
    Sl = args.seq // world
    sl = slice(rank * Sl, (rank + 1) * Sl)
    shards = [b["idx"][:, sl].contiguous() for b in batches]
    labs = [l[:, sl].contiguous() for l in labels]
    from ravnest_amd.ops import FusedAdam
    opt = (FusedAdam(model.parameters(), lr=1e-4) if on_gpu
           else torch.optim.Adam(model.parameters(), lr=1e-4))
    crit = torch.nn.CrossEntropyLoss(ignore_index=-100)
    V = model.cfg.vocab_size

    def step(i):
        opt.zero_grad(set_to_none=True)
        logits = model(shards[i % len(shards)])
        loss = crit(logits.reshape(-1, V).float(),
                    labs[i % len(labs)].reshape(-1))
        loss.backward()
        allreduce_gradients(model)
        opt.step()

    def sync():
        if on_gpu:
            torch.cuda.synchronize(device)

    for i in range(args.warmup):
        step(i)
    sync()
    dist.barrier()
    sync()
    t0 = _time.perf_counter()
    for i in range(args.steps):
        step(i)
    sync()
    elapsed = _time.perf_counter() - t0
    dist.barrier()
    t = torch.tensor([elapsed], dtype=torch.float64,
                     device=device if backend == "nccl" else "cpu")
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())
    if rank == 0:
        samples_per_sec = args.micro_batch * args.steps / elapsed
        print(json.dumps({
            "metric": f"samples/sec (whole node), {args.model} "
                      "sequence-parallel",
            "value": round(samples_per_sec, 2),
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000.0, 2),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": args.dtype if on_gpu else "fp32",
            "data": "synthetic",
            "config": {"model": args.model,
                       "global_batch": args.micro_batch,
                       "seq_len": args.seq,
                       "parallelism": f"sp{world}",
                       "optimizer": "fused_adam" if on_gpu else "adam"},
        }), flush=True)
    dist.barrier()
    dist.destroy_process_group()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=100)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--model", default="bert-base")
    ap.add_argument("--micro-batch", type=int, default=128)
    ap.add_argument("--seq", type=int, default=512)
    ap.add_argument("--reduce-factor", type=int, default=4,
                    help="average DP replicas every N steps")
    ap.add_argument("--parallelism", default="dp",
                    choices=["dp", "pp", "sp"],
                    help="dp: one fused replica per GPU (driver default); "
                         "pp: ONE async pipeline of N stages across the "
                         "N ranks (the reference's headline topology); "
                         "sp: ONE model with the sequence sharded across "
                         "the N ranks (ring attention)")
    ap.add_argument("--dtype", default=None, choices=["bf16", "fp32"])
    ap.add_argument("--cpu", action="store_true",
                    help="CPU sanity mode (tiny model)")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", args.gpus))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if world > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29531")
        # this host driver only supports dmabuf IPC: without it RCCL
        # cross-process CUDA-tensor sharing fails with
        # "hipIpcGetMemHandle: invalid argument"
        os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")

    if args.dtype is None:
        # transformers run bf16-native; ResNet stays fp32 NCHW (measured:
        # MIOpen bf16 convs are ~4x slower than fp32 on this stack, and
        # channels_last is ~15x slower — see BASELINE.md)
        args.dtype = "fp32" if args.model == "resnet50" else "bf16"
    on_gpu = torch.cuda.is_available() and not args.cpu
    if not on_gpu and args.model == "bert-base" and args.cpu:
        args.model = "bert-tiny"
    if args.cpu:
        args.micro_batch = min(args.micro_batch, 4)
        args.seq = min(args.seq, 64)

    # shared-GPU mode: more pipeline ranks than GPUs (e.g. a 2-stage
    # async pipeline exercised on ONE MI355X) — ranks share cuda:0.. and
    # the wire falls back to gloo with CPU staging (comm/p2p.py)
    share_gpu = on_gpu and world > torch.cuda.device_count()
    device = (torch.device("cuda", local_rank % torch.cuda.device_count())
              if on_gpu else torch.device("cpu"))
    if on_gpu:
        torch.cuda.set_device(device)
    set_seed(1234 + 0)  # same init on every replica (DP semantics)

    model, family = build_model(args.model, args.seq)
    model = model.to(device)
    # MI355X-native: bf16 parameters + bf16 compute end to end (no
    # autocast dtype ping-pong); the fused optimizers keep fp32 masters.
    if args.dtype == "bf16" and on_gpu:
        model = model.to(torch.bfloat16)
    if family == "resnet" and on_gpu and \
            os.environ.get("RAVNEST_CHANNELS_LAST", "0") == "1":
        model = model.to(memory_format=torch.channels_last)
    if family == "resnet" and on_gpu:
        # let MIOpen search for fast conv solutions (the warmup steps
        # absorb the find cost; set_seed turned this off for determinism)
        torch.backends.cudnn.benchmark = True
    amp = None

    # ---- synthetic data of the workload's shape (no network: random ids,
    # random MLM labels with 85% ignored), pre-staged on device ----------
    set_seed(1234 + rank)  # different data per replica
    vocab = model.cfg.vocab_size
    nb = 4
    batches, labels = [], []
    for _ in range(nb):
        ids = torch.randint(0, vocab, (args.micro_batch, args.seq),
                            device=device)
        if family == "bert":
            mask = torch.ones(args.micro_batch, args.seq, dtype=torch.int64,
                              device=device)
            y = ids.clone()
            drop = torch.rand(ids.shape, device=device) > 0.15
            y[drop] = -100
            batches.append({"input_ids": ids, "attention_mask": mask})
            labels.append(y)
        elif family == "gpt":
            batches.append({"idx": ids})
            y = torch.roll(ids, -1, dims=1)
            labels.append(y)
        else:  # resnet: TinyImageNet-shaped synthetic images
            X = torch.randn(args.micro_batch, 3, 64, 64, device=device)
            if args.dtype == "bf16" and on_gpu:
                X = X.to(torch.bfloat16)
            if on_gpu and os.environ.get("RAVNEST_CHANNELS_LAST", "0") == "1":
                X = X.to(memory_format=torch.channels_last)
            y = torch.randint(0, 200, (args.micro_batch,), device=device)
            batches.append({"x": X})
            labels.append(y)

    if args.parallelism == "pp" and world > 1:
        run_pp(args, rank, world, device, on_gpu, family, model,
               batches, labels, share_gpu=share_gpu)
        return
    if args.parallelism == "sp" and world > 1:
        run_sp(args, rank, world, device, on_gpu, family, model,
               batches, labels, share_gpu=share_gpu)
        return

    comm = None
    if world > 1:
        comm = CommBackend(rank=rank, world_size=world, edges=[],
                           dp_groups=[list(range(world))], device=device,
                           backend="gloo" if share_gpu else None)

    mi_names = (["input_ids", "attention_mask"] if family == "bert"
                else ["x"] if family == "resnet" else ["idx"])
    cfg = {
        "rank": rank, "world_size": world, "cluster_id": rank, "stage": 0,
        "n_stages": 1, "cluster_length": 1, "stage_ranks": [rank],
        "dp_ranks": list(range(world)), "node_type": "root",
        "model_input_names": mi_names,
    }
    input_template = [{"kind": "model_input", "name": n,
                       "dtype": "torch.int64"} for n in mi_names]
    output_template = {0: {"consumers": [], "final": True,
                           "dtype": "torch.float32"}}

    node = Node(config=cfg, model=model, input_template=input_template,
                output_template=output_template,
                optimizer=FusedAdam, optimizer_params={"lr": 1e-4},
                device=device, criterion=CrossEntropyLoss(-100),
                labels=_ListLoader([labels[i % nb]
                                    for i in range(args.warmup + args.steps + nb)]),
                update_frequency=1, reduce_factor=args.reduce_factor,
                comm=comm, amp_dtype=amp,
                loss_filename="bench_losses.txt")
    node.start()

    def sync():
        if on_gpu:
            torch.cuda.synchronize(device)

    def barrier():
        if comm is not None:
            comm.barrier()

    # ---- warmup ------------------------------------------------------
    for i in range(args.warmup):
        node.forward_compute(**batches[i % nb])
    node.wait_for_backwards()
    sync()
    barrier()
    sync()

    # ---- timed region: EXACTLY K steps -------------------------------
    t0 = time.perf_counter()
    for i in range(args.steps):
        node.forward_compute(**batches[i % nb])
    node.wait_for_backwards()
    sync()
    elapsed = time.perf_counter() - t0
    barrier()

    # max over ranks
    if comm is not None:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if comm.backend == "nccl" else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    global_batch = args.micro_batch * world
    samples_per_sec = global_batch * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        print(json.dumps({
            "metric": "samples/sec (whole node), BERT-base MLM pretraining"
                      if args.model.startswith("bert")
                      else f"samples/sec (whole node), {args.model}",
            "value": round(samples_per_sec, 2),
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype if on_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": global_batch,
                "seq_len": args.seq,
                "parallelism": f"dp{world}",
                "reduce_factor": args.reduce_factor,
                "optimizer": "fused_adam",
            },
        }), flush=True)

    node.stop()
    if comm is not None:
        comm.stop()
        torch.distributed.barrier()
        sys.stdout.flush()
        sys.stderr.flush()
        # gloo/RCCL teardown with live helper threads can abort on exit;
        # results are printed, so leave hard (single-rank keeps the clean
        # exit that rocprof finalization needs)
        os._exit(0)


if __name__ == "__main__":
    main()
