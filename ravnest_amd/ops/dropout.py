"""Dropout with replayable Philox RNG.

The versioned recompute engine replays forwards bit-exactly by restoring
torch RNG state (engine/compute.py; parity reference compute.py:63-68,
227-237). The HIP kernel therefore draws its Philox seed/offset from the
torch HIP generator (advancing it exactly like native kernels do), so
`torch.cuda.set_rng_state` replay reproduces the identical mask —
SURVEY.md section 2.3 "Dropout (with replayable RNG)".
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ._ext import get_ext


class _DropoutFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p):
        ext = get_ext(required=True)
        # seed drawn from the torch CPU generator: the engine's RNG
        # capture/restore (fork_rng) makes the recompute replay identical.
        # The device counter (0 outside graph mode) keeps the seed live
        # across hipGraph replays — see ops/rng.py.
        from . import rng
        seed = int(torch.randint(0, 2**62, (1,)).item())
        y, mask = ext.dropout_fwd(x, p, seed,
                                  rng.device_seed_counter(x.device))
        ctx.save_for_backward(mask)
        ctx.p = p
        return y

    @staticmethod
    def backward(ctx, dy):
        (mask,) = ctx.saved_tensors
        ext = get_ext(required=True)
        dx = ext.dropout_bwd(dy.contiguous(), mask, ctx.p)
        return dx, None


def dropout(x: torch.Tensor, p: float, training: bool = True) -> torch.Tensor:
    if not training or p == 0.0:
        return x
    if x.is_cuda:
        return _DropoutFn.apply(x.contiguous(), p)
    return torch.nn.functional.dropout(x, p, training)


class Dropout(nn.Module):
    _is_leaf_module = True

    def __init__(self, p: float = 0.1):
        super().__init__()
        self.p = p

    def forward(self, x):
        return dropout(x, self.p, self.training)
