"""Fused bias+GELU (tanh approximation) for transformer MLPs.

Reference workload parity: NewGELU (minGPT
model_without_padding_mask.py:25-31), BERT intermediate act — SURVEY.md
section 2.3. The projection GEMM runs on hipBLASLt (plain library GEMM);
the bias-add + GELU epilogue is the fused hand-written HIP kernel
(csrc/gelu.hip) so the intermediate never round-trips through HBM twice.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn

from ._ext import get_ext


def _gelu_tanh_ref(x):
    return 0.5 * x * (1.0 + torch.tanh(
        math.sqrt(2.0 / math.pi) * (x + 0.044715 * x.pow(3))))


class _BiasGeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        ext = get_ext(required=True)
        y = ext.bias_gelu_fwd(x, bias)
        ctx.save_for_backward(x, bias)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, bias = ctx.saved_tensors
        ext = get_ext(required=True)
        if x.dtype == torch.bfloat16:
            # single pass: dx and the bias-grad column sum together
            dx, db = ext.bias_gelu_bwd_db(dy.contiguous(), x, bias)
            return dx, db.to(bias.dtype)
        dx = ext.bias_gelu_bwd(dy.contiguous(), x, bias)
        db = dx.float().reshape(-1, dx.shape[-1]).sum(0).to(bias.dtype)
        return dx, db


def bias_gelu(x: torch.Tensor, bias: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        return _BiasGeluFn.apply(x.contiguous(), bias)
    return _gelu_tanh_ref(x + bias)


def gelu(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        ext = get_ext(required=True)
        return _GeluFn.apply(x.contiguous())
    return _gelu_tanh_ref(x)


class _GeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = get_ext(required=True)
        zero = torch.zeros(x.shape[-1], dtype=x.dtype, device=x.device)
        ctx.save_for_backward(x, zero)
        return ext.bias_gelu_fwd(x, zero)

    @staticmethod
    def backward(ctx, dy):
        x, zero = ctx.saved_tensors
        ext = get_ext(required=True)
        return ext.bias_gelu_bwd(dy.contiguous(), x, zero)


class GELU(nn.Module):
    _is_leaf_module = True

    def forward(self, x):
        return gelu(x)


class LinearGelu(nn.Module):
    """Linear + bias-GELU: hipBLASLt GEMM + fused bias-GELU kernel by
    default; ONE hand-written GEMM with the bias-GELU in the epilogue
    under RAVNEST_HAND_GEMM=1 (ops/linear.py)."""
    _is_leaf_module = True

    def __init__(self, in_features: int, out_features: int):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.zeros(out_features))
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, x):
        from .linear import _HandLinearGeluFn, _hand_ok, hand_gemm_enabled
        if x.is_cuda and hand_gemm_enabled() and _hand_ok(x, self.weight):
            return _HandLinearGeluFn.apply(x, self.weight, self.bias)
        h = torch.nn.functional.linear(x, self.weight)
        return bias_gelu(h, self.bias)


class _SoftmaxFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = get_ext(required=True)
        y = ext.softmax_fwd(x.contiguous())
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        ext = get_ext(required=True)
        return ext.softmax_bwd(dy.contiguous(), y)


def softmax(x: torch.Tensor, dim: int = -1) -> torch.Tensor:
    """Standalone softmax (csrc/softmax.hip) over the last dim; torch
    fallback on CPU or non-last dims. SURVEY.md section 2.3 softmax row."""
    if x.is_cuda and (dim == -1 or dim == x.dim() - 1):
        return _SoftmaxFn.apply(x)
    return torch.softmax(x, dim)


class Softmax(nn.Module):
    _is_leaf_module = True  # fx: device-dependent dispatch

    def __init__(self, dim: int = -1):
        super().__init__()
        self.dim = dim

    def forward(self, x):
        return softmax(x, self.dim)
