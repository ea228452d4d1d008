"""Device-side RNG counter for hipGraph-captured training steps.

The philox dropout kernel XORs a value read from a per-device int64
buffer into its host-drawn seed at RUN time (csrc/dropout.hip). Outside
graph mode the buffer stays 0, so behavior (and the versioned engine's
CPU-RNG replay — reference compute.py:63-68) is unchanged. A captured
step increments the buffer once per replay (the increment is itself part
of the graph), so every replay of the frozen host seed produces a fresh
mask while two same-seed graphed runs stay bit-identical.
"""
from __future__ import annotations

import torch

_counters: dict[int, torch.Tensor] = {}


def device_seed_counter(device: torch.device) -> torch.Tensor:
    """The per-device int64 counter buffer (created on first use)."""
    idx = device.index if device.index is not None else \
        torch.cuda.current_device()
    buf = _counters.get(idx)
    if buf is None:
        buf = torch.zeros(1, dtype=torch.int64, device=f"cuda:{idx}")
        _counters[idx] = buf
    return buf


def bump_seed_counter(device: torch.device) -> None:
    """Advance the counter (device-side add — capturable in a graph)."""
    device_seed_counter(device).add_(1)


def reset_seed_counter(device: torch.device) -> None:
    device_seed_counter(device).zero_()
