"""ComputeEngine unit tests: versioned recompute, RNG replay, grad
accumulation, version GC (the reference's signature mechanism,
SURVEY.md section 2.4)."""
import torch

from ravnest_amd import set_seed
from ravnest_amd.engine.compute import ComputeEngine


def small_model():
    return torch.nn.Sequential(
        torch.nn.Linear(8, 16), torch.nn.ReLU(),
        torch.nn.Dropout(0.5), torch.nn.Linear(16, 4))


def test_forward_capture_and_backward_equivalence():
    """Async recompute-backward on a single stage must equal a plain
    autograd forward+backward when no staleness occurs (in-flight 1)."""
    set_seed(0)
    m1 = small_model()
    set_seed(0)
    m2 = small_model()
    dev = torch.device("cpu")
    opt1 = torch.optim.SGD(m1.parameters(), lr=0.1)
    eng = ComputeEngine(m1, opt1, dev)
    opt2 = torch.optim.SGD(m2.parameters(), lr=0.1)

    set_seed(42)
    x = torch.randn(4, 8)
    g = torch.randn(4, 4)

    # engine path: no-grad forward (captures RNG), then recompute+backward
    set_seed(7)
    outs = eng.forward(0, [x], [True])
    grads, stepped = eng.backward(0, {0: g}, speculative_next=False)
    assert stepped

    # plain path with the same RNG stream
    set_seed(7)
    x2 = x.clone().requires_grad_(True)
    out2 = m2(x2)
    out2.backward(g)
    opt2.step()

    assert torch.allclose(outs[0], out2.detach(), atol=1e-6)
    assert torch.allclose(grads[0], x2.grad, atol=1e-6)
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6)


def test_versioned_staleness():
    """Backward for an old fpid must recompute with the HISTORICAL
    parameter version (activations from old weights)."""
    set_seed(0)
    m = small_model()
    m[2].p = 0.0  # disable dropout for determinism of this check
    dev = torch.device("cpu")
    opt = torch.optim.SGD(m.parameters(), lr=0.5)
    eng = ComputeEngine(m, opt, dev)

    x0, x1 = torch.randn(2, 8), torch.randn(2, 8)
    out0 = eng.forward(0, [x0], [True])
    out1 = eng.forward(1, [x1], [True])
    v_before = eng.current_version
    # backward fpid 0 -> step -> version bump
    eng.backward(0, {0: torch.ones(2, 4)}, speculative_next=False)
    assert eng.current_version == v_before + 1
    # fpid 1 still references old version
    assert eng.fpids[1].version == v_before
    assert v_before in eng.version_to_param
    eng.backward(1, {0: torch.ones(2, 4)}, speculative_next=False)
    # old snapshot released after its last fpid drains
    assert v_before not in eng.version_to_param


def test_rng_replay_dropout():
    """The recompute must replay the EXACT dropout mask of the original
    no-grad forward (parity: fork_rng replay, reference compute.py:234-237)."""
    set_seed(0)
    m = small_model()  # has Dropout(0.5)
    eng = ComputeEngine(m, torch.optim.SGD(m.parameters(), lr=0.0),
                        torch.device("cpu"))
    x = torch.randn(16, 8)
    outs = eng.forward(0, [x], [True])
    eng._recompute(0)
    rec = eng.fpids[0]
    assert torch.allclose(outs[0], rec.recomputed_outputs[0].detach(),
                          atol=1e-6)


def test_update_frequency_accumulation():
    set_seed(0)
    m = small_model()
    m[2].p = 0.0
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    eng = ComputeEngine(m, opt, torch.device("cpu"), update_frequency=2)
    x = torch.randn(2, 8)
    eng.forward(0, [x], [False])
    _, stepped = eng.backward(0, {0: torch.ones(2, 4)},
                              speculative_next=False)
    assert not stepped
    eng.forward(1, [x], [False])
    _, stepped = eng.backward(1, {0: torch.ones(2, 4)},
                              speculative_next=False)
    assert stepped


def test_speculative_recompute():
    set_seed(0)
    m = small_model()
    eng = ComputeEngine(m, torch.optim.SGD(m.parameters(), lr=0.01),
                        torch.device("cpu"))
    x = torch.randn(2, 8)
    eng.forward(0, [x], [True])
    eng.forward(1, [x], [True])
    eng.backward(0, {0: torch.ones(2, 4)}, speculative_next=True)
    eng.join_recompute()
    assert eng.fpids[1].recomputed_outputs is not None
    eng.backward(1, {0: torch.ones(2, 4)}, speculative_next=False)
    assert len(eng.fpids) == 0


def test_find_loss_leaf(tmp_path):
    set_seed(0)
    m = small_model()
    opt = torch.optim.SGD(m.parameters(), lr=0.1)
    eng = ComputeEngine(m, opt, torch.device("cpu"),
                        criterion=lambda out, tgt: torch.nn.functional
                        .mse_loss(out, tgt),
                        loss_filename=str(tmp_path / "losses.txt"))
    x = torch.randn(4, 8)
    y = torch.randn(4, 4)
    grads, stepped, loss = eng.find_loss(0, [x], [True], y)
    assert stepped and loss > 0
    assert grads[0] is not None and grads[0].shape == x.shape
    assert (tmp_path / "losses.txt").exists()


def test_weight_pull_roundtrip():
    set_seed(0)
    m1, m2 = small_model(), small_model()
    e1 = ComputeEngine(m1, None, torch.device("cpu"))
    e2 = ComputeEngine(m2, None, torch.device("cpu"))
    snap = e1.latest_state_snapshot()
    e2.load_param_list(snap["params"])
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.equal(p1, p2)
