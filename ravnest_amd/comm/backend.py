"""Process-group management: one RCCL communicator per pipeline edge plus
per-stage DP groups.

Replaces the reference's per-message insecure gRPC channel creation
(ravnest/communication.py:68,293,302,311) with communicators built once at
startup. Every rank derives the SAME edge list deterministically from the
plan artifacts (or an explicit edge list), because torch.distributed
requires all ranks to call new_group in the same order.
"""
from __future__ import annotations

import datetime
import json
import os
from dataclasses import dataclass
from pathlib import Path

import torch
import torch.distributed as dist

from .p2p import Channel, Message


@dataclass(frozen=True, order=True)
class Edge:
    src: int
    dst: int
    kind: str  # "fwd" | "bwd" | "ctrl"


def edges_from_plan(base_dir: str) -> list[Edge]:
    """Derive the global edge list from node_data plan artifacts.

    fwd edges: producer stage -> each consumer stage (from outputs.json),
    plus root -> any stage consuming a model input (root owns the data
    feed, parity with the reference's model_inputs forwarding,
    communication.py:118-122). bwd edges mirror fwd edges. ctrl edges
    link same-stage ranks of adjacent clusters (latest-weights pull,
    reference communication.py:279-330).
    """
    base = Path(base_dir)
    with open(base / "plan.json") as f:
        plan = json.load(f)
    edges: set[Edge] = set()
    for cl in plan["clusters"]:
        c = cl["cluster_id"]
        ranks = cl["stage_ranks"]
        n_stages = cl["n_stages"]
        for s in range(n_stages):
            with open(base / f"cluster_{c}" / f"stage_{s}" / "outputs.json") as f:
                outs = json.load(f)
            for _, entry in outs.items():
                for cons in entry["consumers"]:
                    j = cons["stage"]
                    if j != s:
                        edges.add(Edge(ranks[s], ranks[j], "fwd"))
                        edges.add(Edge(ranks[j], ranks[s], "bwd"))
            with open(base / f"cluster_{c}" / f"stage_{s}" / "inputs.json") as f:
                ins = json.load(f)
            for src in ins:
                if src.get("kind") == "model_input" and s != 0:
                    edges.add(Edge(ranks[0], ranks[s], "fwd"))
                    edges.add(Edge(ranks[s], ranks[0], "bwd"))
    # ctrl edges between pipeline neighbors (heartbeat / save / recovery)
    for cl in plan["clusters"]:
        ranks = cl["stage_ranks"]
        for a, b in zip(ranks, ranks[1:]):
            edges.add(Edge(a, b, "ctrl"))
            edges.add(Edge(b, a, "ctrl"))
    # ctrl ring across clusters for weight pull / elastic join
    clusters = plan["clusters"]
    if len(clusters) > 1:
        for i, cl in enumerate(clusters):
            nxt = clusters[(i + 1) % len(clusters)]
            for s, r in enumerate(cl["stage_ranks"]):
                if s < len(nxt["stage_ranks"]):
                    edges.add(Edge(r, nxt["stage_ranks"][s], "ctrl"))
                    edges.add(Edge(nxt["stage_ranks"][s], r, "ctrl"))
    # unequal clusters: ctrl edges between every param-range segment's
    # peers (the per-range latest-weights pull targets the MAPPED peer
    # that owns the range in the other cluster — reference
    # node.py:127-135 retrieve_latest_params_data)
    for seg in plan.get("dp_segments", []):
        ranks = seg["ranks"]
        for a in ranks:
            for b in ranks:
                if a != b:
                    edges.add(Edge(a, b, "ctrl"))
    return sorted(edges)


def dp_groups_from_plan(base_dir: str) -> list[list[int]]:
    base = Path(base_dir)
    with open(base / "plan.json") as f:
        plan = json.load(f)
    clusters = plan["clusters"]
    sizes = {c["n_stages"] for c in clusters}
    if len(sizes) != 1:
        return []
    n_stages = sizes.pop()
    return [[c["stage_ranks"][s] for c in clusters] for s in range(n_stages)]


def dp_segments_from_plan(base_dir: str) -> list[dict]:
    """Param-range averaging segments for unequal-cluster DP (parity:
    reference param-range rings, operations/utils.py:463-516)."""
    with open(Path(base_dir) / "plan.json") as f:
        plan = json.load(f)
    return plan.get("dp_segments", [])


class CommBackend:
    """Owns the distributed init, all edge channels and DP groups for one
    rank."""

    def __init__(self, rank: int, world_size: int,
                 edges: list[Edge] | None = None,
                 dp_groups: list[list[int]] | None = None,
                 dp_segments: list[dict] | None = None,
                 base_dir: str | None = None,
                 device: torch.device = torch.device("cpu"),
                 backend: str | None = None,
                 master_addr: str = "127.0.0.1",
                 master_port: int = 29500,
                 timeout_s: int = 300):
        self.rank = rank
        self.world_size = world_size
        self.device = device
        if backend is None:
            backend = "nccl" if device.type == "cuda" else "gloo"
        self.backend = backend

        if not dist.is_initialized():
            os.environ.setdefault("MASTER_ADDR", master_addr)
            os.environ.setdefault("MASTER_PORT", str(master_port))
            if device.type == "cuda":
                torch.cuda.set_device(device)
            dist.init_process_group(
                backend=backend, rank=rank, world_size=world_size,
                timeout=datetime.timedelta(seconds=timeout_s))

        if edges is None and base_dir is not None:
            edges = edges_from_plan(base_dir)
        if dp_groups is None and base_dir is not None:
            dp_groups = dp_groups_from_plan(base_dir)
        if dp_segments is None and base_dir is not None:
            dp_segments = dp_segments_from_plan(base_dir)
        self.edges = edges or []
        self.dp_group_ranks = dp_groups or []

        # every rank creates every group, same order
        self.channels: dict[Edge, Channel] = {}
        self._deliver_cb = None
        for e in self.edges:
            g = dist.new_group([e.src, e.dst]) if e.src != e.dst else None
            ch = Channel(e.src, e.dst, e.kind, g, device, rank)
            self.channels[e] = ch

        self.dp_groups = []
        self.my_dp_group = None
        self.my_dp_ranks: list[int] = [rank]
        for ranks in self.dp_group_ranks:
            g = dist.new_group(ranks)
            self.dp_groups.append((ranks, g))
            if rank in ranks:
                self.my_dp_group = g
                self.my_dp_ranks = ranks

        # param-range segments (unequal clusters): one group per distinct
        # owner-rank tuple, created by EVERY rank in the same order
        self.dp_segments = []          # (ranks, group, param_names)
        self.my_dp_segments = []
        seg_groups: dict[tuple, object] = {}
        for seg in (dp_segments or []):
            t = tuple(seg["ranks"])
            if len(set(t)) <= 1:
                continue
            if t not in seg_groups:
                seg_groups[t] = dist.new_group(sorted(set(t)))
            g = seg_groups[t]
            self.dp_segments.append((list(t), g, seg["params"]))
            if rank in t:
                self.my_dp_segments.append((list(t), g, seg["params"]))

        self._warmed = False

    # -- lifecycle -----------------------------------------------------
    def start(self, deliver):
        """Install the delivery callback and start channel threads.
        `deliver(channel, message)` runs on listener threads."""
        self._deliver_cb = deliver
        self.warmup()
        for ch in self.channels.values():
            ch.deliver = deliver
            ch.start()

    def warmup(self):
        """Force communicator creation deterministically before threads
        start (RCCL communicators are lazily initialized on first use;
        racing first-use from multiple threads is unsafe)."""
        if self._warmed:
            return
        self._warmed = True
        for e in self.edges:
            ch = self.channels[e]
            if ch.group is None:
                continue
            t = torch.zeros(1, device=self.device if self.device.type == "cuda"
                            else "cpu")
            if self.rank == e.src:
                dist.send(t, e.dst, group=ch.group)
            elif self.rank == e.dst:
                dist.recv(t, e.src, group=ch.group)
        for ranks, g in self.dp_groups:
            if self.rank in ranks:
                t = torch.zeros(1, device=self.device if self.device.type == "cuda"
                                else "cpu")
                dist.all_reduce(t, group=g)

    def stop(self, join_timeout: float = 10.0):
        """Graceful shutdown: close every outgoing channel (flush + close
        sentinel), then wait for local channel threads to exit."""
        for ch in self.channels.values():
            if ch.my_rank == ch.src:
                ch.close()
        for ch in self.channels.values():
            ch.join(timeout=join_timeout)

    # -- sending helpers ----------------------------------------------
    def channel(self, src: int, dst: int, kind: str) -> Channel:
        return self.channels[Edge(src, dst, kind)]

    def send(self, dst: int, kind: str, msg: Message):
        self.channel(self.rank, dst, kind).send(msg)

    def barrier(self):
        dist.barrier()
