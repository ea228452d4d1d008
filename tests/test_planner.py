"""Planner tests: fx splitting, routing templates, artifact emission.

Mirrors the parity surface of reference operations/ (SURVEY.md section 3.1).
"""
import json

import pytest
import torch

from ravnest_amd import clusterize, set_seed
from ravnest_amd.models.cnn import CNN
from ravnest_amd.planner import NodeSpec, form_clusters
from ravnest_amd.planner.splitter import (split_model_by_proportions,
                                          hare_niemeyer_counts)


def test_hare_niemeyer():
    assert hare_niemeyer_counts(10, [1, 1, 1]) == [4, 3, 3]
    assert sum(hare_niemeyer_counts(7, [0.5, 0.3, 0.2])) == 7


def test_split_cnn_three_stages():
    set_seed(0)
    m = CNN()
    x = torch.randn(4, 1, 8, 8)
    m.eval()
    res = split_model_by_proportions(m, [1 / 3] * 3, example_args=(x,))
    assert len(res.stages) == 3
    # chain routing
    assert res.stage_inputs[0][0]["kind"] == "model_input"
    assert res.stage_inputs[1][0]["kind"] == "stage"
    assert res.stage_outputs[2][0]["final"]
    # split output == original output
    with torch.no_grad():
        ref = m(x)
        y = res.stages[0](x)
        y = res.stages[1](y)
        y = res.stages[2](y)
    assert torch.allclose(ref, y, atol=1e-6)
    # dtype annotation present for the runtime grad contract
    assert res.stage_inputs[1][0].get("dtype") == "torch.float32"


class SkipNet(torch.nn.Module):
    """Multi-consumer output: stage-0 value consumed by stages 1 AND 2."""

    def __init__(self):
        super().__init__()
        self.a = torch.nn.Linear(16, 16)
        self.b = torch.nn.Linear(16, 16)
        self.c = torch.nn.Linear(16, 16)

    def forward(self, x):
        h = torch.relu(self.a(x))
        g = torch.relu(self.b(h))
        return self.c(g + h)  # skip connection h -> last stage


def test_split_skip_connection():
    set_seed(0)
    m = SkipNet()
    x = torch.randn(2, 16)
    res = split_model_by_proportions(m, [1 / 3] * 3, example_args=(x,))
    # some stage output must have 2 consumers OR h appears as input to two
    # later stages
    consumers = [c for outs in res.stage_outputs
                 for e in outs.values() for c in e["consumers"]]
    assert len(consumers) >= 2
    with torch.no_grad():
        ref = m(x)
        out = res.split_gm(x)
    assert torch.allclose(ref, out, atol=1e-6)


class TupleNet(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.a = torch.nn.Linear(8, 8)
        self.b = torch.nn.Linear(8, 8)

    def forward(self, x):
        h = self.a(x)
        return self.b(h), h.sum()


def test_split_tuple_output():
    m = TupleNet()
    x = torch.randn(2, 8)
    res = split_model_by_proportions(m, [0.5, 0.5], example_args=(x,))
    assert len(res.final_outputs) == 2
    with torch.no_grad():
        r0, r1 = m(x)
        o0, o1 = res.split_gm(x)
    assert torch.allclose(r0, o0, atol=1e-6)
    assert torch.allclose(r1, o1, atol=1e-6)


def test_form_clusters_homogeneous():
    pool = [NodeSpec(name=f"n{i}", ram=100.0, bandwidth=1.0) for i in range(8)]
    # model needs 150 -> 2 nodes per replica -> 4 clusters (max_clusters=5)
    clusters = form_clusters(pool, model_bytes=150.0, seed=1)
    assert all(c.total_ram >= 150.0 for c in clusters)
    assert sum(len(c.nodes) for c in clusters) == 8
    assert len(clusters) >= 2


def test_form_clusters_too_small():
    pool = [NodeSpec(name="n0", ram=1.0)]
    with pytest.raises(ValueError):
        form_clusters(pool, model_bytes=100.0)


def test_clusterize_artifacts(tmp_path):
    set_seed(0)
    m = CNN()
    x = torch.randn(2, 1, 8, 8)
    base = tmp_path / "node_data"
    pool = [NodeSpec(name=f"n{i}", ram=10 * 2**20, bandwidth=1.0)
            for i in range(3)]
    meta = clusterize(m, (x,), node_pool=pool, max_clusters=1,
                      base_dir=str(base))
    assert meta["world_size"] == 3
    for s in range(3):
        d = base / "cluster_0" / f"stage_{s}"
        assert (d / "submod.pt").exists()
        assert (d / "inputs.json").exists()
        assert (d / "outputs.json").exists()
    for r in range(3):
        with open(base / "nodes" / f"node_{r}.json") as f:
            cfg = json.load(f)
        assert cfg["rank"] == r
        assert cfg["node_type"] == ["root", "stem", "leaf"][r]
    # submodels load and chain-forward
    sm = [torch.load(base / "cluster_0" / f"stage_{s}" / "submod.pt",
                     weights_only=False) for s in range(3)]
    with torch.no_grad():
        m.eval()
        [s.eval() for s in sm]
        y = sm[2](sm[1](sm[0](x)))
        ref = m(x)
    assert torch.allclose(ref, y, atol=1e-6)


def test_model_fusion(tmp_path):
    from ravnest_amd import model_fusion
    set_seed(0)
    m = CNN()
    x = torch.randn(2, 1, 8, 8)
    base = tmp_path / "node_data"
    pool = [NodeSpec(name=f"n{i}", ram=10 * 2**20) for i in range(3)]
    clusterize(m, (x,), node_pool=pool, max_clusters=1, base_dir=str(base))
    fused = model_fusion(0, base_dir=str(base), out_dir=str(tmp_path / "out"))
    orig = m.state_dict()
    assert set(fused.keys()) == set(orig.keys())
    for k in orig:
        assert torch.equal(fused[k], orig[k])


def test_form_clusters_heterogeneous():
    """GA placement on an uneven pool (reference genetic.py parity): a
    model needing 10 GiB over {8, 8, 4, 4, 12, 12} GiB nodes — every
    cluster must hold the model, small nodes must pair with big ones,
    and the speed balance must keep clusters within the pool's spread."""
    from ravnest_amd.planner.placement import NodeSpec, form_clusters
    G = 2**30
    pool = [NodeSpec("a", 8 * G, bandwidth=1.0),
            NodeSpec("b", 8 * G, bandwidth=1.0),
            NodeSpec("c", 4 * G, bandwidth=2.0),
            NodeSpec("d", 4 * G, bandwidth=2.0),
            NodeSpec("e", 12 * G, bandwidth=0.5),
            NodeSpec("f", 12 * G, bandwidth=0.5)]
    clusters = form_clusters(pool, model_bytes=10 * G, seed=3)
    assert len(clusters) >= 2  # enough RAM for at least two replicas
    for c in clusters:
        assert c.total_ram >= 10 * G, \
            f"cluster {c.cid} cannot hold the model: {c.total_ram/G} GiB"
    # all nodes placed exactly once
    placed = sorted(n.name for c in clusters for n in c.nodes)
    assert placed == sorted(n.name for n in pool)
    # proportional splits follow member RAM
    big = max(clusters, key=lambda c: len(c.nodes))
    props = big.split_proportions(10 * G)
    assert abs(sum(props) - 1.0) < 1e-6
    rams = [n.ram for n in big.nodes]
    order_by_prop = sorted(range(len(props)), key=lambda i: props[i])
    order_by_ram = sorted(range(len(rams)), key=lambda i: rams[i])
    assert order_by_prop == order_by_ram


def test_form_clusters_insufficient_ram():
    from ravnest_amd.planner.placement import NodeSpec, form_clusters
    import pytest as _pytest
    with _pytest.raises(ValueError):
        form_clusters([NodeSpec("a", 2**30)], model_bytes=10 * 2**30)


def test_load_node_pool_reference_format(tmp_path):
    """Reference-compatible node_configs.json round trip (parity:
    spawn_node_pool load_from_configs, operations/utils.py:24-50)."""
    import json
    from ravnest_amd.planner.placement import load_node_pool, mi355x_pool
    cfg = {"0": {"IP": "0.0.0.0:8080", "benchmarks": {"ram": 8,
                                                      "bandwidth": 10}},
           "1": {"IP": "0.0.0.0:8081", "benchmarks": {"ram": 16,
                                                      "bandwidth": 5}},
           "10": {"IP": "0.0.0.0:8090"}}  # defaults + numeric sort
    p = tmp_path / "node_configs.json"
    p.write_text(json.dumps(cfg))
    pool = load_node_pool(p)
    assert [n.name for n in pool] == ["node_0", "node_1", "node_10"]
    assert pool[0].ram == 8 * 2**30 and pool[0].bandwidth == 10
    assert pool[1].address == "0.0.0.0:8081"
    assert pool[2].ram == 8 * 2**30  # default
    gpus = mi355x_pool(8)
    assert len(gpus) == 8 and gpus[3].device_index == 3
    assert gpus[0].ram > 250 * 2**30  # 288 GB minus reserve
