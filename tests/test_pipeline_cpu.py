"""End-to-end async pipeline on CPU: 3 stages, 3 processes, gloo backend.

This is the reference's canonical walkthrough workload (CNN on synthetic
8x8 digits, 3-node pipeline — SURVEY.md section 7 minimum slice) run over
the MI355X framework's comm stack with the gloo backend standing in for
RCCL. Asserts: training runs, loss decreases, shutdown is clean.
"""
import os

import numpy as np
import torch
import torch.multiprocessing as mp

from ravnest_amd import clusterize, set_seed
from ravnest_amd.models.cnn import CNN
from ravnest_amd.planner import NodeSpec


def _make_loaders(seed=42, n=256, batch=32):
    g = torch.Generator()
    g.manual_seed(seed)
    rng = np.random.RandomState(0)
    X = rng.rand(n, 1, 8, 8).astype("float32")
    y = rng.randint(0, 10, size=n)
    Y = np.zeros((n, 10), dtype="float32")
    Y[np.arange(n), y] = 1.0
    ds = list(zip(torch.tensor(X), torch.tensor(Y)))
    from torch.utils.data import DataLoader
    return DataLoader(ds, batch_size=batch, shuffle=True, generator=g)


def _loss_fn(preds, targets):
    return torch.nn.functional.mse_loss(preds, targets[1])


def _worker(rank, world, base_dir, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["GLOO_SOCKET_IFNAME"] = os.environ.get("GLOO_SOCKET_IFNAME", "lo")
    os.chdir(out_dir)
    set_seed(42)
    from ravnest_amd import Node, Trainer

    train_loader = _make_loaders()
    node = Node(name=f"node_{rank}", base_dir=base_dir,
                optimizer=torch.optim.Adam,
                device=torch.device("cpu"),
                criterion=_loss_fn,
                labels=_make_loaders(),  # same seed => same order as root
                update_frequency=1)
    node.start()
    trainer = Trainer(node=node, train_loader=train_loader,
                      epochs=3, batch_size=32, inputs_dtype=torch.float32)
    if rank == 0:
        trainer.train()
        node.stop_cluster()
    else:
        trainer.prelim_checks()
    node.stop()


def test_three_stage_pipeline_cpu(tmp_path):
    set_seed(42)
    model = CNN()
    x = torch.randn(2, 1, 8, 8)
    base = str(tmp_path / "node_data")
    pool = [NodeSpec(name=f"n{i}", ram=10 * 2**20) for i in range(3)]
    clusterize(model, (x,), node_pool=pool, max_clusters=1, base_dir=base)

    port = 29510 + (os.getpid() % 500)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker,
                         args=(r, 3, base, port, str(tmp_path)))
             for r in range(3)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    for p in procs:
        assert p.exitcode == 0, f"worker exited {p.exitcode}"

    # leaf (rank 2) wrote losses.txt in tmp_path (cwd of workers)
    loss_file = tmp_path / "losses.txt"
    assert loss_file.exists(), "leaf produced no losses.txt"
    losses = [float(l) for l in loss_file.read_text().split()]
    n_batches = 256 // 32
    assert len(losses) == 3 * n_batches
    first_epoch = sum(losses[:n_batches])
    last_epoch = sum(losses[-n_batches:])
    assert last_epoch < first_epoch, \
        f"loss did not decrease: {first_epoch} -> {last_epoch}"


def _comp_worker(rank, base_dir, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["GLOO_SOCKET_IFNAME"] = os.environ.get("GLOO_SOCKET_IFNAME", "lo")
    os.chdir(out_dir)
    set_seed(42)
    from ravnest_amd import Node, Trainer
    node = Node(name=f"node_{rank}", base_dir=base_dir,
                optimizer=torch.optim.Adam,
                device=torch.device("cpu"),
                criterion=_loss_fn,
                labels=_make_loaders(),
                update_frequency=1,
                compression=True)  # bf16 wire for activations AND grads
    node.start()
    trainer = Trainer(node=node, train_loader=_make_loaders(),
                      epochs=3, batch_size=32, inputs_dtype=torch.float32)
    if rank == 0:
        trainer.train()
        node.stop_cluster()
    else:
        trainer.prelim_checks()
    node.stop()


def test_pipeline_wire_compression(tmp_path):
    """bf16 on-the-wire compression for both directions (parity:
    reference fp16 compression, utils.py:184-194; grads cast back to the
    output dtype at the consumer)."""
    set_seed(42)
    model = CNN()
    x = torch.randn(2, 1, 8, 8)
    base = str(tmp_path / "node_data")
    pool = [NodeSpec(name=f"n{i}", ram=10 * 2**20) for i in range(2)]
    clusterize(model, (x,), node_pool=pool, max_clusters=1, base_dir=base)
    port = 29900 + (os.getpid() % 90)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_comp_worker,
                         args=(r, base, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    for p in procs:
        assert p.exitcode == 0, f"worker exited {p.exitcode}"
    losses = [float(l) for l in (tmp_path / "losses.txt").read_text().split()]
    n_batches = 256 // 32
    assert len(losses) == 3 * n_batches
    assert sum(losses[-n_batches:]) < sum(losses[:n_batches])


def _shared_loader_worker(rank, base_dir, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(out_dir)
    set_seed(42)
    from ravnest_amd import Node, Trainer
    # the walkthrough pattern: ONE loader object wraps both the node's
    # label feed and the trainer's input feed
    loader = _make_digits_loader()
    node = Node(name=f"node_{rank}", base_dir=base_dir,
                optimizer=torch.optim.Adam,
                device=torch.device("cpu"),
                criterion=_loss_fn,
                labels=loader,
                update_frequency=1)
    node.start()
    trainer = Trainer(node=node, train_loader=loader, epochs=12,
                      batch_size=64, inputs_dtype=torch.float32)
    if rank == 0:
        trainer.train()
        node.stop_cluster()
    else:
        trainer.prelim_checks()
    node.stop()


def _make_digits_loader(seed=42):
    from sklearn import datasets
    data = datasets.load_digits()
    y = np.zeros((len(data.target), 10), dtype="float32")
    y[np.arange(len(y)), data.target] = 1.0
    X = data.data.reshape(-1, 1, 8, 8).astype("float32")[:1024]
    g = torch.Generator()
    g.manual_seed(seed)
    from torch.utils.data import DataLoader
    return DataLoader(list(zip(torch.tensor(X), torch.tensor(y[:1024]))),
                      generator=g, shuffle=True, batch_size=64)


def test_shared_loader_label_alignment(tmp_path):
    """Regression: a SHUFFLING loader shared between node.labels and the
    trainer must keep the root's data order aligned with the leaf's
    labels. The eager iter() in the label iterator used to consume one
    generator draw at Node init, desynchronizing the two sides — the
    model then learned only the output mean (loss floor ~0.09 on
    one-hot digits). With alignment, real learning takes loss far below
    that floor."""
    set_seed(42)
    model = CNN()
    base = str(tmp_path / "node_data")
    pool = [NodeSpec(name=f"n{i}", ram=10 * 2**20) for i in range(3)]
    clusterize(model, (torch.randn(2, 1, 8, 8),), node_pool=pool,
               max_clusters=1, base_dir=base)
    port = 29450 + (os.getpid() % 50)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_shared_loader_worker,
                         args=(r, base, port, str(tmp_path)))
             for r in range(3)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
    for p in procs:
        assert p.exitcode == 0, f"worker exited {p.exitcode}"
    losses = [float(x) for x in (tmp_path / "losses.txt").read_text().split()]
    n = 1024 // 64
    last_epoch = sum(losses[-n:]) / n
    assert last_epoch < 0.03, \
        f"data/label misalignment suspected: last-epoch loss {last_epoch}"


def _f8_worker(rank, base_dir, port, out_dir):
    os.environ["RAVNEST_WIRE_FP8"] = "1"
    _comp_worker(rank, base_dir, port, out_dir)


def test_pipeline_fp8_wire(tmp_path):
    """Full 2-stage pipeline with the fp8 wire (RAVNEST_WIRE_FP8=1) on
    top of bf16 compression: activations AND grads ship as scaled-e4m3
    bytes (comm/p2p.py). Training must still converge on the CNN
    workload — exercises the engine-level decompression path against
    channel-level fp8 reconstruction."""
    set_seed(42)
    model = CNN()
    x = torch.randn(2, 1, 8, 8)
    base = str(tmp_path / "node_data")
    pool = [NodeSpec(name=f"n{i}", ram=10 * 2**20) for i in range(2)]
    clusterize(model, (x,), node_pool=pool, max_clusters=1, base_dir=base)
    port = 29700 + (os.getpid() % 90)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_f8_worker,
                         args=(r, base, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
    for p in procs:
        assert p.exitcode == 0, f"worker exited {p.exitcode}"
    losses = [float(l) for l in (tmp_path / "losses.txt").read_text().split()]
    n_batches = 256 // 32
    assert len(losses) == 3 * n_batches
    assert sum(losses[-n_batches:]) < sum(losses[:n_batches])
