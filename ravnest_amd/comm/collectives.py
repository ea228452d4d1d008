"""Bucketed parameter / optimizer-state averaging over RCCL.

Replaces the reference's hand-rolled chunked ring all-reduce over gRPC
(ravnest/communication.py:125-277 `parallel_ring_reduce`/`single_ring_reduce`
with per-iteration counter polling, endpoints.py:91-143): on MI355X the
same-stage DP replicas live on one xGMI-connected node, so parameter
averaging is a bucketed `all_reduce` on the per-stage RCCL communicator.
xGMI is point-to-point (7 links x ~153 GB/s per GPU) so ring collectives
are per-link bound: buckets are sized large (default 64 MiB) to amortize
launch/latency, and the whole pass can run on a side stream so it overlaps
backward compute.
"""
from __future__ import annotations

import torch
import torch.distributed as dist


def _all_reduce_avg(flat: torch.Tensor, group, world: int):
    if dist.get_backend(group) == "nccl" and hasattr(dist.ReduceOp, "AVG"):
        dist.all_reduce(flat, op=dist.ReduceOp.AVG, group=group)
    else:
        dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=group)
        flat.div_(world)


def average_tensors(tensors: list[torch.Tensor], group,
                    bucket_bytes: int = 64 * 2**20) -> None:
    """Average a list of tensors in place across `group`, flattened into
    large buckets (one collective per bucket, not per tensor)."""
    if group is None:
        return
    world = dist.get_world_size(group=group)
    if world <= 1:
        return
    bucket: list[torch.Tensor] = []
    size = 0

    def flush():
        nonlocal bucket, size
        if not bucket:
            return
        flat = torch.cat([t.detach().reshape(-1) for t in bucket])
        _all_reduce_avg(flat, group, world)
        off = 0
        for t in bucket:
            n = t.numel()
            t.detach().copy_(flat[off:off + n].view_as(t))
            off += n
        bucket, size = [], 0

    # group same-dtype tensors per bucket
    by_dtype: dict[torch.dtype, list[torch.Tensor]] = {}
    for t in tensors:
        by_dtype.setdefault(t.dtype, []).append(t)
    for _, ts in by_dtype.items():
        for t in ts:
            bucket.append(t)
            size += t.numel() * t.element_size()
            if size >= bucket_bytes:
                flush()
        flush()


def _master_tensors(optimizer, params) -> list[torch.Tensor]:
    """fp32 master copies the fused optimizers keep for bf16 params.
    These MUST be averaged whenever the params are: the optimizers read
    the master (not the bf16 param) on every step, so averaging only the
    bf16 values would be silently discarded at the next step."""
    if optimizer is None:
        return []
    out = []
    for p in params:
        st = optimizer.state.get(p)
        if st:
            m = st.get("master")
            if isinstance(m, torch.Tensor):
                out.append(m)
    return out


def average_parameters(model: torch.nn.Module, group,
                       bucket_bytes: int = 64 * 2**20,
                       optimizer=None) -> None:
    params = [p for p in model.parameters()]
    average_tensors(params + _master_tensors(optimizer, params), group,
                    bucket_bytes)


def average_optimizer_state(optimizer: torch.optim.Optimizer, group,
                            bucket_bytes: int = 64 * 2**20) -> None:
    """Average floating-point optimizer state tensors (Adam moments, LAMB
    trust state, SGD momentum) across the DP group (parity: reference
    `average_optim` path, communication.py:131-138,176-179,267-272).
    fp32 masters are excluded: they ride with average_parameters."""
    tensors = []
    for st in optimizer.state.values():
        for k, v in st.items():
            if k == "master":
                continue
            if isinstance(v, torch.Tensor) and v.is_floating_point() and v.numel() > 0:
                tensors.append(v)
    average_tensors(tensors, group, bucket_bytes)


class AsyncReducer:
    """Overlapped DP parameter averaging (SURVEY.md §7 "Overlap
    engineering" — the reference simply blocks, communication.py:125-277).

    launch(): snapshot the tensors into per-dtype flat buffers and start
    the averaging all_reduce on a SIDE stream (GPU) or a background
    thread (CPU/gloo), so the pipeline's next microbatch overlaps the
    collective. join_into(): make the compute stream wait on the
    collective (GPU-side wait, no host sync) and copy the averaged
    values back.

    Semantics note: between launch and join the local params are frozen
    (no optimizer step happens inside a reduce window when
    reduce_threshold is a multiple of update_frequency), but the next
    microbatch's gradients are computed at the PRE-average weights; the
    averaged values land just before the next optimizer step. The
    synchronous path (`average_parameters`) remains for exact-parity
    call sites."""

    def __init__(self, group, device: torch.device):
        self.group = group
        self.device = device
        self._stream = (torch.cuda.Stream(device)
                        if device.type == "cuda" else None)
        self._pending = None
        self._thread = None

    @property
    def pending(self) -> bool:
        return self._pending is not None

    def launch(self, tensors: list[torch.Tensor]) -> None:
        assert self._pending is None, "previous async reduce not joined"
        world = dist.get_world_size(group=self.group)
        if world <= 1 or not tensors:
            return
        by_dtype: dict[torch.dtype, list[torch.Tensor]] = {}
        for t in tensors:
            by_dtype.setdefault(t.dtype, []).append(t)
        if self._stream is not None:
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream(self.device))
            flats = []
            with torch.cuda.stream(self._stream):
                self._stream.wait_event(ev)
                for ts in by_dtype.values():
                    flat = torch.cat([t.detach().reshape(-1) for t in ts])
                    _all_reduce_avg(flat, self.group, world)
                    flats.append((ts, flat))
            done = torch.cuda.Event()
            done.record(self._stream)
            self._pending = (flats, done)
        else:
            flats = []
            for ts in by_dtype.values():
                flats.append((ts, torch.cat(
                    [t.detach().reshape(-1) for t in ts])))

            self._failed = False

            def run():
                try:
                    for _, flat in flats:
                        _all_reduce_avg(flat, self.group, world)
                except Exception as e:  # peer died mid-collective
                    self._failed = True
                    print(f"[AsyncReducer] collective failed: {e}",
                          flush=True)

            import threading
            self._thread = threading.Thread(target=run, daemon=True)
            self._thread.start()
            self._pending = (flats, None)

    def join_into(self) -> bool:
        """Install the averaged values; returns True if a reduce landed."""
        if self._pending is None:
            return False
        flats, done = self._pending
        if self._stream is not None:
            # compute stream waits for the collective — no host sync
            torch.cuda.current_stream(self.device).wait_event(done)
        elif self._thread is not None:
            self._thread.join()
            self._thread = None
            if getattr(self, "_failed", False):
                # peer died mid-collective: DISCARD the partial buffers
                # (never install garbage); the health monitor suspends
                # future averaging
                self._pending = None
                self._failed = False
                return False
        for ts, flat in flats:
            off = 0
            for t in ts:
                n = t.numel()
                t.detach().copy_(flat[off:off + n].view_as(t))
                off += n
            if self._stream is not None:
                flat.record_stream(torch.cuda.current_stream(self.device))
        self._pending = None
        return True


def average_parameter_segments(model: torch.nn.Module, segments,
                               optimizer=None, average_optim: bool = False,
                               bucket_bytes: int = 64 * 2**20) -> None:
    """Unequal-cluster DP: average each param-range segment on its own
    group (parity: reference param-range rings,
    operations/utils.py:463-516). `segments` = [(ranks, group, names)]
    restricted to this rank. fp32 masters always ride with their params;
    the rest of the optimizer state (moments / momentum) only when
    `average_optim`."""
    named = dict(model.named_parameters())
    for _, group, names in segments:
        params = [named[n] for n in names if n in named]
        if not params:
            continue
        average_tensors(params + _master_tensors(optimizer, params), group,
                        bucket_bytes)
        if optimizer is not None and average_optim:
            sts = []
            for p in params:
                st = optimizer.state.get(p)
                if not st:
                    continue
                for k, v in st.items():
                    if k == "master":
                        continue
                    if isinstance(v, torch.Tensor) and \
                            v.is_floating_point() and v.numel() > 0:
                        sts.append(v)
            if sts:
                average_tensors(sts, group, bucket_bytes)
