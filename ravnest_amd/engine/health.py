"""Health monitoring + fault injection.

The reference ships an unused Ping RPC and no heartbeat (SURVEY.md
section 5: a dead peer hangs its poll loops). Here: a heartbeat thread
pings every ctrl-connected peer over the ctrl channels and records
last-seen timestamps / RTTs; `Node.health()` exposes them and
`on_peer_lost` fires when a peer goes silent past the timeout. Recovery
pairs with `Node.update_with_latest_weights()` (the reference's
latest-weights pull, communication.py:279-330).

Fault injection (tests / chaos drills): FaultInjector can drop or delay
a node's outgoing messages, or freeze its dispatch loop, to exercise the
monitor and the pull-based recovery.
"""
from __future__ import annotations

import random
import threading
import time


class HealthMonitor:
    def __init__(self, node, interval: float = 2.0, timeout: float = 10.0,
                 on_peer_lost=None, on_peer_recovered=None):
        self.node = node
        self.interval = interval
        self.timeout = timeout
        self.on_peer_lost = on_peer_lost
        self.on_peer_recovered = on_peer_recovered
        self.last_seen: dict[int, float] = {}
        self.rtt: dict[int, float] = {}
        self._ping_sent: dict[int, float] = {}
        self._lost: set[int] = set()
        self._thread: threading.Thread | None = None
        self._stop = threading.Event()

    def peers(self) -> list[int]:
        if self.node.comm is None:
            return []
        return sorted({e.dst for e, ch in self.node.comm.channels.items()
                       if e.kind == "ctrl" and e.src == self.node.rank})

    def start(self):
        if self._thread is not None:
            return
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name=f"health:rank{self.node.rank}")
        self._thread.start()

    def stop(self):
        self._stop.set()

    def note_pong(self, src: int):
        now = time.monotonic()
        self.last_seen[src] = now
        if src in self._ping_sent:
            self.rtt[src] = now - self._ping_sent[src]
        if src in self._lost:
            self._lost.discard(src)
            if self.on_peer_recovered is not None:
                self.on_peer_recovered(src)

    def note_ping(self, src: int):
        self.last_seen[src] = time.monotonic()

    def _loop(self):
        from ..comm import Message
        from ..strings import ActionTypes
        while not self._stop.is_set() and not self.node._stop.is_set():
            now = time.monotonic()
            for peer in self.peers():
                try:
                    self._ping_sent[peer] = now
                    self.node.comm.send(peer, "ctrl", Message(
                        action=ActionTypes.STOP, fpid=-2, tensors=[],
                        extra=3))  # extra=3: PING
                except Exception:
                    pass  # a dead peer's channel may refuse the send
                seen = self.last_seen.get(peer)
                if seen is not None and now - seen > self.timeout and \
                        peer not in self._lost:
                    self._lost.add(peer)
                    if self.on_peer_lost is not None:
                        self.on_peer_lost(peer)
            self._stop.wait(self.interval)

    def health(self) -> dict:
        now = time.monotonic()
        return {
            "peers": {p: {"last_seen_s": (now - self.last_seen[p])
                          if p in self.last_seen else None,
                          "rtt_s": self.rtt.get(p),
                          "lost": p in self._lost}
                      for p in self.peers()},
        }


class FaultInjector:
    """Wraps a node's comm send path to drop/delay messages, for
    fault-tolerance tests."""

    def __init__(self, node, drop_prob: float = 0.0, delay_s: float = 0.0,
                 seed: int = 0):
        self.node = node
        self.drop_prob = drop_prob
        self.delay_s = delay_s
        self.rng = random.Random(seed)
        self._orig_send = None

    def install(self):
        comm = self.node.comm
        if comm is None:
            return
        self._orig_send = comm.send

        def send(dst, kind, msg):
            if self.rng.random() < self.drop_prob:
                return  # dropped
            if self.delay_s:
                time.sleep(self.delay_s)
            self._orig_send(dst, kind, msg)

        comm.send = send

    def remove(self):
        if self._orig_send is not None:
            self.node.comm.send = self._orig_send
            self._orig_send = None
