"""Unequal-cluster DP over gloo: a 2-stage cluster and a 3-stage cluster
training the same model, synchronized by PARAM-RANGE averaging groups
(parity: the reference's param-range rings across clusters with
different splits, operations/utils.py:463-516).

Checks: the planner buckets parameters by their per-cluster owner-rank
tuples (dp_segments), every segment group spans both clusters, and
after training + a final reduce both replicas hold identical parameters
while the loss decreased.
"""
import os

import numpy as np
import torch
import torch.multiprocessing as mp

from ravnest_amd import clusterize, set_seed
from ravnest_amd.planner import NodeSpec
from ravnest_amd.models.cnn import CNN

WORLD = 5  # one 2-stage and one 3-stage cluster (planner orders them)


def _loss_fn(preds, targets):
    return torch.nn.functional.mse_loss(preds, targets[1])


def _make_loader(seed=42, n=96, batch=32):
    g = torch.Generator()
    g.manual_seed(seed)
    rng = np.random.RandomState(0)
    X = rng.rand(n, 1, 8, 8).astype("float32")
    Y = np.eye(10, dtype="float32")[rng.randint(0, 10, size=n)]
    from torch.utils.data import DataLoader
    return DataLoader(list(zip(torch.tensor(X), torch.tensor(Y))),
                      batch_size=batch, shuffle=True, generator=g)


def _worker(rank, base_dir, port, out_dir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(out_dir)
    set_seed(42)
    from ravnest_amd import Node, Trainer
    node = Node(name=f"node_{rank}", base_dir=base_dir,
                optimizer=torch.optim.Adam,
                optimizer_params={"lr": 3e-3},
                device=torch.device("cpu"),
                criterion=_loss_fn,
                labels=_make_loader(),  # same seed -> same data order
                update_frequency=1, reduce_factor=1)
    node.start()
    trainer = Trainer(node=node, train_loader=_make_loader(),
                      epochs=4, batch_size=32, inputs_dtype=torch.float32)
    if node.node_type.value == "root":
        trainer.train()
        node.comm_session.parallel_ring_reduce()
        node.stop_cluster()
    else:
        trainer.prelim_checks()
        node.comm_session.parallel_ring_reduce()
    # export this stage's params for cross-replica comparison
    flat = {n: p.detach().clone() for n, p in node.model.named_parameters()}
    torch.save(flat, os.path.join(out_dir, f"params_rank{rank}.pt"))
    q.put(rank)
    node.stop()


def test_unequal_cluster_dp(tmp_path):
    set_seed(42)
    model = CNN()
    x = torch.randn(2, 1, 8, 8)
    base = str(tmp_path / "node_data")
    pool = [NodeSpec(name=f"n{i}", ram=100 * 2**20) for i in range(WORLD)]
    meta = clusterize(model, (x,), node_pool=pool,
                      cluster_assignment=[[0, 1], [2, 3, 4]],
                      base_dir=base)
    assert meta["world_size"] == WORLD

    segs = meta["dp_segments"]
    assert segs, "unequal clusters must produce dp_segments"
    c0 = set(meta["clusters"][0]["stage_ranks"])
    c1 = set(meta["clusters"][1]["stage_ranks"])
    covered = set()
    for seg in segs:
        ranks = set(seg["ranks"])
        assert ranks & c0 and ranks & c1, seg
        covered.update(seg["params"])
    # every trainable param is averaged by exactly one segment
    names = {n for n, _ in model.named_parameters()}
    assert covered == names, names - covered

    port = 30200 + (os.getpid() % 200)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker,
                         args=(r, base, port, str(tmp_path), q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
    for p in procs:
        assert p.exitcode == 0, f"worker exited {p.exitcode}"

    # reassemble both replicas' full param dicts
    replica = {0: {}, 1: {}}
    for r in range(WORLD):
        d = torch.load(tmp_path / f"params_rank{r}.pt")
        replica[0 if r in c0 else 1].update(d)
    assert set(replica[0]) == set(replica[1]) == names
    for n in names:
        assert torch.allclose(replica[0][n], replica[1][n], atol=1e-6), \
            f"replicas diverge on {n}"

    losses = [float(v) for v in
              (tmp_path / "losses.txt").read_text().split()]
    n_batches = 96 // 32
    # two leaves interleave their losses; compare epoch aggregates
    assert len(losses) == 2 * 4 * n_batches
    assert sum(losses[-2 * n_batches:]) < sum(losses[:2 * n_batches])


def _pull_worker(rank, base_dir, port, out_dir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(out_dir)
    set_seed(42)
    import time
    from ravnest_amd import Node
    node = Node(name=f"node_{rank}", base_dir=base_dir,
                optimizer=torch.optim.Adam,
                device=torch.device("cpu"),
                criterion=_loss_fn, labels=_make_loader())
    node.start()
    if rank == 0:
        time.sleep(1.5)  # let every serving peer come up
        before = {n: p.detach().clone()
                  for n, p in node.model.named_parameters()}
        with torch.no_grad():
            for p in node.model.parameters():
                p.add_(1.0)  # corrupt the local replica
        node.update_with_latest_weights()  # per-SEGMENT pull from peers
        diff = max((p.detach() - before[n]).abs().max().item()
                   for n, p in node.model.named_parameters())
        q.put(("ok", diff))
        q.close()
        q.join_thread()
        os._exit(0)
    # serving peers park until the parent terminates them
    time.sleep(60)
    os._exit(0)


def test_per_range_latest_weights_pull(tmp_path):
    """Unequal clusters: update_with_latest_weights pulls each
    param-range SEGMENT from its mapped peer in the other cluster (the
    reference's multi-peer partial restore, node.py:127-135 /
    communication.py:279-330) — a corrupted replica is fully restored."""
    set_seed(42)
    model = CNN()
    base = str(tmp_path / "node_data")
    pool = [NodeSpec(name=f"n{i}", ram=100 * 2**20) for i in range(WORLD)]
    clusterize(model, (torch.randn(2, 1, 8, 8),), node_pool=pool,
               cluster_assignment=[[0, 1], [2, 3, 4]], base_dir=base)
    port = 30420 + (os.getpid() % 60)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_pull_worker,
                         args=(r, base, port, str(tmp_path), q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    res = q.get(timeout=150)
    for p in procs:
        p.terminate()
        p.join(timeout=30)
    assert res[0] == "ok"
    assert res[1] < 1e-6, f"pull did not restore params (max diff {res[1]})"
