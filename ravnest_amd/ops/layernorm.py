"""Fused LayerNorm (bf16/fp32 in, fp32 statistics) on CDNA4.

Reference workload parity: LayerNorm appears in every transformer example
(minGPT model_without_padding_mask.py:121-124, BERT) — SURVEY.md section
2.3 op table. GPU path: hand-written HIP kernel (csrc/layernorm.hip), one
workgroup per row, vectorized bf16x8 loads, wave-level reductions. CPU
path: torch native (tests compare the HIP kernel against fp32 torch).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ._ext import get_ext


class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        ext = get_ext(required=True)
        y, mean, rstd = ext.layernorm_fwd(x, weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        ext = get_ext(required=True)
        dx, dw, db = ext.layernorm_bwd(dy.contiguous(), x, weight, mean, rstd)
        return dx, dw, db, None


def layer_norm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
               eps: float = 1e-12) -> torch.Tensor:
    if x.is_cuda:
        return _LayerNormFn.apply(x.contiguous(), weight, bias, eps)
    return F.layer_norm(x, (x.shape[-1],), weight, bias, eps)


class FusedLayerNorm(nn.Module):
    _is_leaf_module = True  # fx: custom autograd inside

    def __init__(self, hidden: int, eps: float = 1e-12):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden))
        self.bias = nn.Parameter(torch.zeros(hidden))
        self.eps = eps

    def forward(self, x):
        return layer_norm(x, self.weight, self.bias, self.eps)


class _AddLayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b, weight, bias, eps):
        ext = get_ext(required=True)
        y, s, mean, rstd = ext.layernorm_add_fwd(a, b, weight, bias, eps)
        ctx.save_for_backward(s, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        s, weight, mean, rstd = ctx.saved_tensors
        ext = get_ext(required=True)
        dx, dw, db = ext.layernorm_bwd(dy.contiguous(), s, weight, mean,
                                       rstd)
        # d/da == d/db, but the two returns must be DISTINCT objects:
        # autograd's in-place accumulation fast path can steal a grad
        # tensor for one accumulator and add_ into it later, which would
        # double-count an aliased twin
        return dx, dx.clone(), dw, db, None


def add_layer_norm(a, b, weight, bias, eps: float = 1e-12):
    """y = LayerNorm(a + b) with the residual add fused into the LN read
    pass (one HBM round-trip instead of two)."""
    if a.is_cuda:
        if b.dtype != a.dtype:  # mixed autocast inputs: follow the residual
            b = b.to(a.dtype)
        return _AddLayerNormFn.apply(a.contiguous(), b.contiguous(),
                                     weight, bias, eps)
    return F.layer_norm(a + b, (a.shape[-1],), weight, bias, eps)


class AddLayerNorm(nn.Module):
    _is_leaf_module = True

    def __init__(self, hidden: int, eps: float = 1e-12):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden))
        self.bias = nn.Parameter(torch.zeros(hidden))
        self.eps = eps

    def forward(self, a, b):
        return add_layer_norm(a, b, self.weight, self.bias, self.eps)
