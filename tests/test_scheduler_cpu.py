"""Per-node lr scheduler + val_freq parity (reference node.py:213-215,
517-518,585-586,603-604 and trainer.py:21-22).

The scheduler is built INSIDE Node so stem/leaf ranks step it too: the
leaf detects the epoch boundary (label iterator wrap) and the flag rides
the backward messages so every upstream rank steps in the same epoch.
"""
import os
import time

import numpy as np
import torch
import torch.multiprocessing as mp

from ravnest_amd import Node, Trainer, clusterize, set_seed
from ravnest_amd.models.cnn import CNN
from ravnest_amd.planner import NodeSpec


def _make_loader(seed=42, n=128, batch=32):
    g = torch.Generator()
    g.manual_seed(seed)
    rng = np.random.RandomState(0)
    X = rng.rand(n, 1, 8, 8).astype("float32")
    Y = np.eye(10, dtype="float32")[rng.randint(0, 10, size=n)]
    from torch.utils.data import DataLoader
    return DataLoader(list(zip(torch.tensor(X), torch.tensor(Y))),
                      batch_size=batch, shuffle=True, generator=g)


def _loss_fn(preds, targets):
    return torch.nn.functional.mse_loss(preds, targets[1])


def _sched_worker(rank, base_dir, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["GLOO_SOCKET_IFNAME"] = os.environ.get("GLOO_SOCKET_IFNAME",
                                                      "lo")
    os.chdir(out_dir)
    set_seed(42)
    node = Node(name=f"node_{rank}", base_dir=base_dir,
                optimizer=torch.optim.Adam,
                optimizer_params={"lr": 1e-3},
                lr_scheduler=torch.optim.lr_scheduler.StepLR,
                lr_scheduler_params={"step_size": 1, "gamma": 0.5},
                device=torch.device("cpu"),
                criterion=_loss_fn,
                labels=_make_loader(),
                update_frequency=1)
    node.start()
    trainer = Trainer(node=node, train_loader=_make_loader(), epochs=3,
                      batch_size=32, inputs_dtype=torch.float32)
    if rank == 0:
        trainer.train()
        node.stop_cluster()
    else:
        trainer.prelim_checks()
    with open(f"lr_rank{rank}.txt", "w") as f:
        f.write(repr(node.optimizer.param_groups[0]["lr"]))
    node.stop()


def test_per_node_scheduler_steps_on_every_rank(tmp_path):
    """A 3-stage pipeline where the STEM's lr must decay too — the
    reference builds the scheduler per node exactly so passive ranks are
    not stuck at the initial lr."""
    set_seed(42)
    base = str(tmp_path / "node_data")
    pool = [NodeSpec(name=f"n{i}", ram=10 * 2**20) for i in range(3)]
    clusterize(CNN(), (torch.randn(2, 1, 8, 8),), node_pool=pool,
               max_clusters=1, base_dir=base)
    port = 29720 + (os.getpid() % 90)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_sched_worker,
                         args=(r, base, port, str(tmp_path)))
             for r in range(3)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    for p in procs:
        assert p.exitcode == 0, f"worker exited {p.exitcode}"
    # 3 epochs => 2 observed epoch wraps => lr = 1e-3 * 0.5^2 on EVERY
    # rank (the wrap is only seen when the next epoch's first label is
    # drawn, so the final epoch's boundary is not counted)
    for r in range(3):
        lr = float((tmp_path / f"lr_rank{r}.txt").read_text())
        assert abs(lr - 1e-3 * 0.25) < 1e-12, f"rank {r} lr={lr}"


def _mk_fused_node(tmpdir):
    set_seed(0)
    model = CNN()
    cfg = {"rank": 0, "world_size": 1, "cluster_id": 0, "stage": 0,
           "n_stages": 1, "cluster_length": 1, "stage_ranks": [0],
           "dp_ranks": [0], "node_type": "root",
           "model_input_names": ["x"],
           "template_path": str(tmpdir) + "/"}
    labels = [(torch.randn(8, 1, 8, 8),
               torch.eye(10)[torch.randint(0, 10, (8,))])
              for _ in range(8)]
    test_labels = [(torch.randn(8, 1, 8, 8),
                    torch.eye(10)[torch.randint(0, 10, (8,))])
                   for _ in range(2)]
    return Node(config=cfg, model=model,
                input_template=[{"kind": "model_input", "name": "x",
                                 "dtype": "torch.float32"}],
                output_template={0: {"consumers": [], "final": True,
                                     "dtype": "torch.float32"}},
                optimizer=torch.optim.Adam, optimizer_params={"lr": 1e-3},
                criterion=lambda p, t: torch.nn.functional.mse_loss(p, t[1]),
                labels=labels, test_labels=test_labels,
                device=torch.device("cpu"),
                loss_filename=str(tmpdir / "losses.txt"))


def _wait_for(fn, timeout=20.0):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if fn():
            return True
        time.sleep(0.05)
    return fn()


def test_val_freq_batch_cadence(tmp_path, monkeypatch):
    """val_freq > 1 validates every val_freq training batches (the
    reference documents this cadence, trainer.py:21-22)."""
    monkeypatch.chdir(tmp_path)
    node = _mk_fused_node(tmp_path)
    node.start()
    train = [(torch.randn(8, 1, 8, 8), None) for _ in range(8)]
    val = [(torch.randn(8, 1, 8, 8), None) for _ in range(2)]
    trainer = Trainer(node=node, train_loader=train, val_loader=val,
                      val_freq=4, epochs=2, batch_size=8,
                      inputs_dtype=torch.float32)
    trainer.train()
    # 16 forwards, cadence 4 => 4 validation passes x 2 val batches
    assert _wait_for(lambda: len(node.val_accuracies) == 8), \
        f"expected 8 val entries, got {len(node.val_accuracies)}"
    node.stop()


def test_val_freq_default_per_epoch(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    node = _mk_fused_node(tmp_path)
    node.start()
    train = [(torch.randn(8, 1, 8, 8), None) for _ in range(8)]
    val = [(torch.randn(8, 1, 8, 8), None) for _ in range(2)]
    trainer = Trainer(node=node, train_loader=train, val_loader=val,
                      val_freq=1, epochs=2, batch_size=8,
                      inputs_dtype=torch.float32)
    trainer.train()
    # per-epoch validation: 2 epochs x 2 val batches
    assert _wait_for(lambda: len(node.val_accuracies) == 4), \
        f"expected 4 val entries, got {len(node.val_accuracies)}"
    node.stop()


def test_scheduler_per_step_mode(tmp_path, monkeypatch):
    """lr_step_on_epoch_change=False steps the schedule after every
    optimizer step (reference node.py:529-531)."""
    monkeypatch.chdir(tmp_path)
    set_seed(0)
    model = CNN()
    cfg = {"rank": 0, "world_size": 1, "cluster_id": 0, "stage": 0,
           "n_stages": 1, "cluster_length": 1, "stage_ranks": [0],
           "dp_ranks": [0], "node_type": "root",
           "model_input_names": ["x"],
           "template_path": str(tmp_path) + "/"}
    labels = [(torch.randn(8, 1, 8, 8),
               torch.eye(10)[torch.randint(0, 10, (8,))])
              for _ in range(4)]
    node = Node(config=cfg, model=model,
                input_template=[{"kind": "model_input", "name": "x",
                                 "dtype": "torch.float32"}],
                output_template={0: {"consumers": [], "final": True,
                                     "dtype": "torch.float32"}},
                optimizer=torch.optim.Adam, optimizer_params={"lr": 1e-3},
                lr_scheduler=torch.optim.lr_scheduler.StepLR,
                lr_scheduler_params={"step_size": 1, "gamma": 0.5},
                lr_step_on_epoch_change=False,
                criterion=lambda p, t: torch.nn.functional.mse_loss(p, t[1]),
                labels=labels, device=torch.device("cpu"),
                loss_filename=str(tmp_path / "losses.txt"))
    node.start()
    for _ in range(3):
        node.forward_compute(tensors=torch.randn(8, 1, 8, 8))
    node.wait_for_backwards()
    assert _wait_for(
        lambda: abs(node.optimizer.param_groups[0]["lr"]
                    - 1e-3 * 0.5 ** 3) < 1e-12)
    node.stop()
