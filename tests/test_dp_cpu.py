"""Data-parallel tests over gloo: fused single-stage DP replicas with
periodic parameter averaging, and hybrid 2-stage PP x 2-replica DP.

Parity surface: the reference's cross-cluster ring averaging
(communication.py:125-277) as a bucketed all_reduce, and the hybrid
PP x DP topology of SURVEY.md section 2.2.
"""
import os

import numpy as np
import torch
import torch.multiprocessing as mp

from ravnest_amd import clusterize, set_seed
from ravnest_amd.planner import NodeSpec
from ravnest_amd.models.cnn import CNN


def _loss_fn(preds, targets):
    return torch.nn.functional.mse_loss(preds, targets[1])


def _make_loader(seed=42, n=128, batch=32):
    g = torch.Generator()
    g.manual_seed(seed)
    rng = np.random.RandomState(0)
    X = rng.rand(n, 1, 8, 8).astype("float32")
    Y = np.eye(10, dtype="float32")[rng.randint(0, 10, size=n)]
    from torch.utils.data import DataLoader
    return DataLoader(list(zip(torch.tensor(X), torch.tensor(Y))),
                      batch_size=batch, shuffle=True, generator=g)


def _fused_dp_worker(rank, base_dir, port, out_dir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(out_dir)
    set_seed(42)
    from ravnest_amd import Node
    node = Node(name=f"node_{rank}", base_dir=base_dir,
                optimizer=torch.optim.Adam,
                device=torch.device("cpu"),
                criterion=_loss_fn,
                labels=_make_loader(),
                update_frequency=1, reduce_factor=4)
    node.start()
    loader = _make_loader()
    for _ in range(2):
        for X, _y in loader:
            node.forward_compute(tensors=X)
    node.wait_for_backwards()
    node.comm_session.parallel_ring_reduce()
    # after the final averaging all replicas must hold identical params
    flat = torch.cat([p.detach().reshape(-1) for p in node.model.parameters()])
    q.put((rank, flat.sum().item(), flat[:5].tolist()))
    node.stop()


def test_fused_dp_two_replicas(tmp_path):
    """2 clusters x 1 stage: every rank is a fused root+leaf; periodic
    parameter averaging keeps replicas in sync."""
    set_seed(42)
    model = CNN()
    x = torch.randn(2, 1, 8, 8)
    base = str(tmp_path / "node_data")
    # two 1-node clusters: each node fits the model alone
    pool = [NodeSpec(name=f"n{i}", ram=100 * 2**20) for i in range(2)]
    meta = clusterize(model, (x,), node_pool=pool, max_clusters=2,
                      base_dir=base)
    assert meta["n_clusters"] == 2

    port = 29600 + (os.getpid() % 300)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_fused_dp_worker,
                         args=(r, base, port, str(tmp_path), q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for p in procs:
        assert p.exitcode == 0
    (r0, s0, head0), (r1, s1, head1) = sorted(results)
    assert abs(s0 - s1) < 1e-4, f"replicas diverged after averaging: {s0} vs {s1}"
    assert np.allclose(head0, head1, atol=1e-6)


def _pp_dp_worker(rank, base_dir, port, out_dir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(out_dir)
    set_seed(42)
    from ravnest_amd import Node, Trainer
    node = Node(name=f"node_{rank}", base_dir=base_dir,
                optimizer=torch.optim.Adam,
                device=torch.device("cpu"),
                criterion=_loss_fn,
                labels=_make_loader(),
                update_frequency=1, reduce_factor=4)
    node.start()
    trainer = Trainer(node=node, train_loader=_make_loader(), epochs=2,
                      batch_size=32, inputs_dtype=torch.float32)
    if node.node_type.value == "root" or node.fused:
        trainer.train()
        node.stop_cluster()
    else:
        trainer.prelim_checks()
    flat = torch.cat([p.detach().reshape(-1) for p in node.model.parameters()])
    q.put((rank, node.stage, flat.sum().item()))
    node.stop()


def test_pp_dp_hybrid(tmp_path):
    """2 clusters x 2 stages (4 ranks): async PP inside each cluster, DP
    averaging across clusters per stage."""
    set_seed(42)
    model = CNN()
    x = torch.randn(2, 1, 8, 8)
    base = str(tmp_path / "node_data")
    bytes_needed = sum(p.numel() * p.element_size()
                       for p in model.parameters()) * 6
    # each node holds ~60% of the model -> 2 nodes per replica
    pool = [NodeSpec(name=f"n{i}", ram=int(bytes_needed * 0.6))
            for i in range(4)]
    meta = clusterize(model, (x,), node_pool=pool, max_clusters=4,
                      base_dir=base)
    assert meta["n_clusters"] == 2
    assert meta["world_size"] == 4

    port = 29700 + (os.getpid() % 300)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_pp_dp_worker,
                         args=(r, base, port, str(tmp_path), q))
             for r in range(4)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(4):
        rank, stage, ssum = q.get(timeout=240)
        results[rank] = (stage, ssum)
    for p in procs:
        p.join(timeout=60)
    for p in procs:
        assert p.exitcode == 0
    # same-stage replicas hold identical params after the final averaging
    by_stage = {}
    for rank, (stage, ssum) in results.items():
        by_stage.setdefault(stage, []).append(ssum)
    for stage, sums in by_stage.items():
        assert len(sums) == 2
        assert abs(sums[0] - sums[1]) < 1e-3, \
            f"stage {stage} replicas diverged: {sums}"


def _pull_worker(rank, base_dir, port, out_dir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(out_dir)
    set_seed(42)
    from ravnest_amd import Node
    node = Node(name=f"node_{rank}", base_dir=base_dir,
                optimizer=torch.optim.Adam,
                device=torch.device("cpu"),
                criterion=_loss_fn,
                labels=_make_loader())
    node.start()
    if rank == 1:
        # simulate a lagging/fresh replica: perturb, then pull the peer's
        # latest snapshot over the ctrl channel (reference latest-weights
        # pull, communication.py:279-330)
        with torch.no_grad():
            for p in node.model.parameters():
                p.add_(1.0)
        node.update_with_latest_weights(src_rank=0)
    import time
    time.sleep(1.0)  # rank 0 serves the request via its dispatch thread
    flat = torch.cat([p.detach().reshape(-1) for p in node.model.parameters()])
    q.put((rank, flat.sum().item()))
    node.stop()


def test_latest_weights_pull(tmp_path):
    """Elastic-join weight pull between DP replicas over ctrl channels."""
    set_seed(42)
    model = CNN()
    x = torch.randn(2, 1, 8, 8)
    base = str(tmp_path / "node_data")
    pool = [NodeSpec(name=f"n{i}", ram=100 * 2**20) for i in range(2)]
    clusterize(model, (x,), node_pool=pool, max_clusters=2, base_dir=base)
    port = 29750 + (os.getpid() % 90)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_pull_worker,
                         args=(r, base, port, str(tmp_path), q))
             for r in range(2)]
    for p in procs:
        p.start()
    res = dict(q.get(timeout=120) for _ in range(2))
    for p in procs:
        p.join(timeout=60)
    assert abs(res[0] - res[1]) < 1e-4, f"pull failed: {res}"
