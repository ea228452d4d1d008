"""Cluster formation: group accelerator/host nodes into DP replicas.

Capability parity with the reference's GA-based planner
(ravnest/operations/genetic.py:6-70 `genetic_algorithm`/`calculate_fitness`
and operations/utils.py:24-106 spawn_node_pool / cluster_formation /
calculate_split_percentages). On a homogeneous single 8×MI355X node the GA
degenerates to an even grid, but the fitness-based interface is kept so
heterogeneous pools (mixed GPU memory / host RAM) still place correctly.

Fitness per candidate grouping (lower is better):
  ram_penalty  = sum over clusters of max(0, model_bytes - cluster_ram)
  balance_pen  = (max cluster speed - min cluster speed)
  fitness      = RAM_WEIGHT * ram_penalty_norm + balance_pen
"""
from __future__ import annotations

import json
import random
from dataclasses import dataclass, field
from pathlib import Path


@dataclass
class NodeSpec:
    name: str
    ram: float  # bytes available for model state on this device
    bandwidth: float = 1.0  # relative speed score
    address: str | None = None  # informational; runtime uses ranks
    device_index: int | None = None

    @property
    def speed(self) -> float:
        return self.bandwidth


@dataclass
class Cluster:
    cid: int
    nodes: list[NodeSpec] = field(default_factory=list)

    @property
    def total_ram(self) -> float:
        return sum(n.ram for n in self.nodes)

    @property
    def speed(self) -> float:
        return min(n.bandwidth for n in self.nodes) if self.nodes else 0.0

    def split_proportions(self, model_bytes: float) -> list[float]:
        """Per-stage parameter share proportional to each member's RAM
        (parity: reference calculate_split_percentages,
        operations/utils.py:92-106)."""
        total = self.total_ram
        if total <= 0:
            return [1.0 / len(self.nodes)] * len(self.nodes)
        return [n.ram / total for n in self.nodes]


RAM_WEIGHT = 100.0


def _fitness(genome: list[int], pool: list[NodeSpec], n_clusters: int,
             model_bytes: float) -> float:
    rams = [0.0] * n_clusters
    speeds: list[list[float]] = [[] for _ in range(n_clusters)]
    for node_idx, cid in enumerate(genome):
        rams[cid] += pool[node_idx].ram
        speeds[cid].append(pool[node_idx].speed)
    penalty = 0.0
    cluster_speed = []
    for cid in range(n_clusters):
        if not speeds[cid]:
            return float("inf")  # empty cluster is invalid
        penalty += max(0.0, model_bytes - rams[cid]) / max(model_bytes, 1.0)
        cluster_speed.append(sum(speeds[cid]))
    return RAM_WEIGHT * penalty + (max(cluster_speed) - min(cluster_speed))


def form_clusters(pool: list[NodeSpec], model_bytes: float,
                  max_clusters: int = 5, pop: int = 120, gens: int = 150,
                  seed: int = 0) -> list[Cluster]:
    """Search over (n_clusters, assignment) with a small GA; returns the
    best valid grouping. Falls back to one big cluster when nothing fits."""
    rng = random.Random(seed)
    total_ram = sum(n.ram for n in pool)
    if total_ram < model_bytes:
        raise ValueError(
            f"node pool RAM {total_ram/2**30:.1f} GiB cannot hold the model "
            f"({model_bytes/2**30:.1f} GiB)")

    best_genome, best_fit, best_k = None, float("inf"), 1
    upper = min(max_clusters, len(pool))
    for k in range(1, upper + 1):
        if total_ram / k < model_bytes:
            continue  # k replicas can't each hold the model
        population = [[rng.randrange(k) for _ in pool] for _ in range(pop)]
        # Seed with the round-robin assignment (optimal for homogeneous pools)
        population[0] = [i % k for i in range(len(pool))]
        for _ in range(gens):
            scored = sorted(population,
                            key=lambda g: _fitness(g, pool, k, model_bytes))
            if _fitness(scored[0], pool, k, model_bytes) == 0.0:
                break
            nxt = scored[: pop // 5]  # elitism
            while len(nxt) < pop:
                a, b = rng.sample(scored[: pop // 2], 2)
                cut = rng.randrange(1, len(pool)) if len(pool) > 1 else 0
                child = a[:cut] + b[cut:]
                if rng.random() < 0.05:
                    child[rng.randrange(len(pool))] = rng.randrange(k)
                nxt.append(child)
            population = nxt
        genome = min(population, key=lambda g: _fitness(g, pool, k, model_bytes))
        fit = _fitness(genome, pool, k, model_bytes)
        # prefer MORE clusters (more DP) when fitness is equally feasible
        if fit < best_fit or (fit == best_fit and k > best_k and fit < RAM_WEIGHT):
            best_genome, best_fit, best_k = genome, fit, k

    clusters = [Cluster(cid=c) for c in range(best_k)]
    for node_idx, cid in enumerate(best_genome):
        clusters[cid].nodes.append(pool[node_idx])
    return [c for c in clusters if c.nodes]


def load_node_pool(config_path: str | Path) -> list[NodeSpec]:
    """Read the reference-compatible node_configs.json
    ({"0": {"IP": ..., "benchmarks": {"ram": GB, "bandwidth": Mbps}}}).
    Parity: reference spawn_node_pool(mode='load_from_configs'),
    operations/utils.py:24-50 (ram is given in GB there; scaled to bytes)."""
    with open(config_path) as f:
        cfg = json.load(f)
    pool = []
    for key in sorted(cfg, key=lambda s: int(s) if str(s).isdigit() else s):
        ent = cfg[key]
        bench = ent.get("benchmarks", {})
        pool.append(NodeSpec(
            name=f"node_{key}",
            ram=float(bench.get("ram", 8)) * 2**30,
            bandwidth=float(bench.get("bandwidth", 1.0)),
            address=ent.get("IP"),
        ))
    return pool


def mi355x_pool(n_gpus: int, hbm_bytes: float = 288 * 2**30 * 0.9) -> list[NodeSpec]:
    """Homogeneous pool for one MI355X node: n_gpus devices, 288 GB HBM3E
    each (10% reserved for activations/comm buffers)."""
    return [NodeSpec(name=f"node_{i}", ram=hbm_bytes, bandwidth=1.0,
                     device_index=i) for i in range(n_gpus)]
