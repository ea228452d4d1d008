"""Pipeline-runtime coverage for the remaining model families on CPU:
GPT (causal LM through a 3-stage pipeline) and Inception-V3 (multi-branch
routing — SURVEY.md section 7 hard part "test with Inception"). Plus the
evaluate/val_accuracy path on the CNN walkthrough."""
import os

import numpy as np
import torch
import torch.multiprocessing as mp

from ravnest_amd import clusterize, set_seed
from ravnest_amd.planner import NodeSpec


def _run_cluster(nproc, base, port, tmp, worker):
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=worker, args=(r, base, port, str(tmp)))
             for r in range(nproc)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=420)
    for p in procs:
        assert p.exitcode == 0, f"worker exited {p.exitcode}"


# ---------------- GPT 3-stage pipeline ----------------
def _gpt_data():
    g = np.random.RandomState(0)
    X = g.randint(0, 16, size=(48, 11))
    return torch.tensor(X, dtype=torch.int64)


def _gpt_loss(preds, targets):
    return torch.nn.functional.cross_entropy(
        preds.reshape(-1, preds.size(-1)).float(), targets.reshape(-1))


def _gpt_worker(rank, base, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(out_dir)
    set_seed(0)
    from ravnest_amd import Node, Trainer
    X = _gpt_data()
    batches = [X[i * 8:(i + 1) * 8] for i in range(6)]
    node = Node(name=f"node_{rank}", base_dir=base,
                optimizer=torch.optim.Adam, optimizer_params={"lr": 1e-3},
                device=torch.device("cpu"), criterion=_gpt_loss,
                labels=batches * 20)
    node.start()
    if node.node_type.value == "root":
        for epoch in range(2):
            for b in batches:
                node.forward_compute(tensors=b)
        node.wait_for_backwards(timeout=300)
        node.stop_cluster()
    else:
        Trainer(node=node).prelim_checks()
    node.stop()


def test_gpt_pipeline_3stage(tmp_path):
    from ravnest_amd.models import GPT, GPTConfig
    set_seed(0)
    model = GPT(GPTConfig.nano(vocab_size=16, block_size=11))
    base = str(tmp_path / "node_data")
    pool = [NodeSpec(name=f"n{i}", ram=10 * 2**20) for i in range(3)]
    clusterize(model, (_gpt_data()[:2],), node_pool=pool, max_clusters=1,
               base_dir=base)
    port = 30100 + (os.getpid() % 90)
    _run_cluster(3, base, port, tmp_path, _gpt_worker)
    losses = [float(x) for x in
              (tmp_path / "losses.txt").read_text().split()]
    assert len(losses) == 12


# ---------------- Inception 2-stage (multi-branch) ----------------
def _inc_worker(rank, base, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(out_dir)
    set_seed(0)
    from ravnest_amd import Node, Trainer
    g = torch.Generator().manual_seed(1)
    batches = [(torch.randn(2, 3, 32, 32, generator=g),
                torch.randint(0, 10, (2,), generator=g)) for _ in range(2)]
    node = Node(name=f"node_{rank}", base_dir=base,
                optimizer=torch.optim.SGD, optimizer_params={"lr": 0.01},
                device=torch.device("cpu"),
                criterion=lambda p, t: torch.nn.functional.cross_entropy(
                    p.float(), t[1]),
                labels=batches * 10)
    node.start()
    if node.node_type.value == "root":
        for X, _ in batches:
            node.forward_compute(tensors=X)
        node.wait_for_backwards(timeout=300)
        node.stop_cluster()
    else:
        Trainer(node=node).prelim_checks()
    node.stop()


def test_inception_pipeline_2stage(tmp_path):
    from ravnest_amd.models import Inception3
    set_seed(0)
    model = Inception3()
    base = str(tmp_path / "node_data")
    pool = [NodeSpec(name=f"n{i}", ram=400 * 2**20) for i in range(2)]
    clusterize(model, (torch.randn(2, 3, 32, 32),), node_pool=pool,
               max_clusters=1, base_dir=base)
    port = 30200 + (os.getpid() % 90)
    _run_cluster(2, base, port, tmp_path, _inc_worker)
    losses = [float(x) for x in
              (tmp_path / "losses.txt").read_text().split()]
    assert len(losses) == 2 and all(np.isfinite(losses))


# ---------------- CNN evaluate/val path ----------------
def _cnn_val_worker(rank, base, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(out_dir)
    set_seed(42)
    from ravnest_amd import Node, Trainer
    from ravnest_amd.models.cnn import CNN  # noqa: F401
    g = np.random.RandomState(0)
    X = torch.tensor(g.rand(64, 1, 8, 8), dtype=torch.float32)
    Y = torch.tensor(np.eye(10)[g.randint(0, 10, 64)], dtype=torch.float32)
    train = [(X[i * 16:(i + 1) * 16], Y[i * 16:(i + 1) * 16])
             for i in range(4)]
    node = Node(name=f"node_{rank}", base_dir=base,
                optimizer=torch.optim.Adam, device=torch.device("cpu"),
                criterion=lambda p, t: torch.nn.functional.mse_loss(p, t[1]),
                labels=train * 10, test_labels=train * 10)
    node.start()
    trainer = Trainer(node=node, train_loader=train, val_loader=train,
                      epochs=1, inputs_dtype=torch.float32)
    if node.node_type.value == "root":
        trainer.train()
        # prediction output type: leaf saves prediction.pt
        # (parity: reference node.py:683 prediction handler)
        node.no_grad_forward_compute(tensors=X[:8],
                                     output_type="prediction")
        # Trainer.pred uses output_type='accuracy' (exact reference
        # parity: trainer.py:112 passes 'accuracy')
        trainer.pred(X[:8].numpy())
        node.wait_for_backwards()
        import time as _t
        deadline = _t.time() + 60
        while not os.path.exists("prediction.pt") and _t.time() < deadline:
            _t.sleep(0.2)
        node.stop_cluster()
    else:
        trainer.prelim_checks()
    node.stop()


def test_cnn_pipeline_evaluate(tmp_path):
    from ravnest_amd.models.cnn import CNN
    set_seed(42)
    base = str(tmp_path / "node_data")
    pool = [NodeSpec(name=f"n{i}", ram=10 * 2**20) for i in range(3)]
    clusterize(CNN(), (torch.randn(2, 1, 8, 8),), node_pool=pool,
               max_clusters=1, base_dir=base)
    port = 30300 + (os.getpid() % 90)
    _run_cluster(3, base, port, tmp_path, _cnn_val_worker)
    # leaf wrote val accuracies during trainer.train()'s val pass
    vf = tmp_path / "val_accuracies.txt"
    assert vf.exists(), "no val_accuracies.txt from the evaluate path"
    pred = torch.load(tmp_path / "prediction.pt")
    assert pred.shape == (8, 10), pred.shape
    accs = [float(x) for x in vf.read_text().split()]
    assert len(accs) == 4
