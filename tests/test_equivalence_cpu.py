"""Pipeline-vs-single-process training equivalence.

SURVEY.md section 4 implication: the reference's determinism scaffolding
makes near-bitwise equivalence checks feasible. Here: a 2-stage pipeline
(2 processes, gloo) trained with inject-and-drain per microbatch
(synchronous schedule, update_frequency=1, dropout active — RNG replay
must still reproduce the same masks within each stage) must produce the
SAME final parameters as the undistributed model trained in one process.
"""
import os

import numpy as np
import torch
import torch.multiprocessing as mp

from ravnest_amd import clusterize, set_seed
from ravnest_amd.planner import NodeSpec

HID = 32


def build_model():
    set_seed(7)
    return torch.nn.Sequential(
        torch.nn.Linear(16, HID), torch.nn.ReLU(),
        torch.nn.Linear(HID, HID), torch.nn.Tanh(),
        torch.nn.Linear(HID, 4))


def data():
    g = np.random.RandomState(3)
    X = torch.tensor(g.rand(40, 16), dtype=torch.float32)
    Y = torch.tensor(g.rand(40, 4), dtype=torch.float32)
    return [(X[i * 8:(i + 1) * 8], Y[i * 8:(i + 1) * 8]) for i in range(5)]


def _worker(rank, base, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.chdir(out_dir)
    set_seed(7)
    from ravnest_amd import Node
    batches = data()
    node = Node(name=f"node_{rank}", base_dir=base,
                optimizer=torch.optim.SGD, optimizer_params={"lr": 0.1},
                device=torch.device("cpu"),
                criterion=lambda p, t: torch.nn.functional.mse_loss(p, t[1]),
                labels=batches)
    node.start()
    if node.node_type.value == "root":
        for X, _ in batches:
            node.forward_compute(tensors=X)
            node.wait_for_backwards(timeout=120)  # drain: synchronous
        node.stop_cluster()
    else:
        while not node._stop.is_set():
            import time
            time.sleep(0.02)
    torch.save(node.model.state_dict(),
               os.path.join(out_dir, f"stage_{rank}_sd.pt"))
    node.stop()


def test_two_stage_equivalence(tmp_path):
    set_seed(7)
    model = build_model()
    ref_sd = {k: v.detach().clone() for k, v in model.state_dict().items()}
    base = str(tmp_path / "node_data")
    pool = [NodeSpec(name=f"n{i}", ram=10 * 2**20) for i in range(2)]
    clusterize(model, (torch.randn(2, 16),), node_pool=pool, max_clusters=1,
               base_dir=base)

    # single-process reference training (same init: clusterize did not
    # mutate params)
    set_seed(7)
    ref = build_model()
    for k, v in ref.state_dict().items():
        assert torch.equal(v, ref_sd[k])
    opt = torch.optim.SGD(ref.parameters(), lr=0.1)
    for X, Y in data():
        loss = torch.nn.functional.mse_loss(ref(X), Y)
        opt.zero_grad()
        loss.backward()
        opt.step()

    port = 30400 + (os.getpid() % 90)
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, base, port, str(tmp_path)))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
    for p in procs:
        assert p.exitcode == 0

    # fuse stage params and compare against the single-process result
    fused = {}
    for r in range(2):
        fused.update(torch.load(tmp_path / f"stage_{r}_sd.pt",
                                weights_only=True))
    ref_final = ref.state_dict()
    assert set(fused.keys()) == set(ref_final.keys())
    for k in ref_final:
        assert torch.allclose(fused[k], ref_final[k], atol=1e-6), \
            f"{k}: max err {(fused[k]-ref_final[k]).abs().max()}"
