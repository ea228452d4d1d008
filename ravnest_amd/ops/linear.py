"""Linear projections: library GEMMs by default, hand-written CDNA4 MFMA
GEMM (csrc/gemm.hip — 256^2 glds-staged phase schedule, fused bias/GELU
epilogues) behind RAVNEST_HAND_GEMM=1.

Default path: hipBLASLt (TunableOp-selected algorithms). The hand kernel
measures 0.7-0.8x hipBLASLt on the BERT-base projection shapes
(profiles/r02_gemm_vs_blaslt.txt), so the library stays the default for
the headline bench; the hand path is complete (fwd + dgrad via W^T,
wgrad via library, fused bias-GELU storing the pre-activation) and
numerics-tested, and the flag flips every projection in a model built
through make_linear()/LinearGelu.

NOTE: the transformer models route plain projections through
make_linear() — nn.Linear by default: the explicit backward wins
isolated per-op timings (tools/bench_linear_bwd.py) but loses end-to-end
inside the captured step (see make_linear docstring), so autograd's
addmm backward stays the default. RAVNEST_EXPLICIT_LINEAR=1 flips it.
"""
from __future__ import annotations

import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from ._ext import get_ext


def hand_gemm_enabled() -> bool:
    return os.environ.get("RAVNEST_HAND_GEMM", "0") == "1"


def _hand_ok(x: torch.Tensor, w: torch.Tensor) -> bool:
    """Shapes the hand kernel handles: bf16, K % 64 == 0, K >= 128,
    operands < 4 GiB (32-bit staging offsets)."""
    return (x.is_cuda and x.dtype == torch.bfloat16
            and w.dtype == torch.bfloat16
            and w.shape[1] % 64 == 0 and w.shape[1] >= 128
            and x.numel() * 2 < 2**32 and w.numel() * 2 < 2**32)


def _hand_wgrad(dy2: torch.Tensor, x2: torch.Tensor) -> torch.Tensor:
    """dW = dy^T @ x via the split-K tr16 kernel where it measured
    faster than the library (profiles/r02_gemm_vs_blaslt.txt: wins the
    N*K <= 2304*768 shapes, loses the wide ones)."""
    M, N = dy2.shape
    K = x2.shape[1]
    if (M % 64 == 0 and M >= 128 and N >= 8 and K >= 8
            and N * K <= 2304 * 768):
        ext = get_ext(required=True)
        return ext.gemm_wgrad_bf16(dy2, x2).to(dy2.dtype)
    return dy2.t() @ x2


def _hand_dgrad(dy2: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """dx = dy @ W: reduction over N -> needs W^T as the kernel's
    [rows][k] operand; falls back to the library when N is ragged."""
    N = w.shape[0]
    if N % 64 == 0 and N >= 128 and dy2.numel() * 2 < 2**32:
        ext = get_ext(required=True)
        wt = w.t().contiguous()
        (dx2,) = ext.gemm_nt_bf16(dy2.contiguous(), wt, None, 0)
        return dx2
    return dy2 @ w


class _HandLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b):
        ext = get_ext(required=True)
        x2 = x.reshape(-1, x.shape[-1]).contiguous()
        if b is not None:
            (y,) = ext.gemm_nt_bf16(x2, w, b.contiguous(), 1)
        else:
            (y,) = ext.gemm_nt_bf16(x2, w, None, 0)
        ctx.save_for_backward(x2, w)
        ctx.has_bias = b is not None
        ctx.x_shape = x.shape
        return y.reshape(*x.shape[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        x2, w = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx = _hand_dgrad(dy2, w).reshape(ctx.x_shape)
        dw = _hand_wgrad(dy2, x2)
        db = None
        if ctx.has_bias:
            ext = get_ext(required=True)
            db = ext.colsum_bf16(dy2).to(dy.dtype)
        return dx, dw, db


class _HandLinearGeluFn(torch.autograd.Function):
    """y = gelu(x @ W^T + b) in ONE kernel (the epilogue also stores the
    post-bias pre-activation h for the exact backward)."""

    @staticmethod
    def forward(ctx, x, w, b):
        ext = get_ext(required=True)
        x2 = x.reshape(-1, x.shape[-1]).contiguous()
        y, h = ext.gemm_nt_bf16(x2, w, b.contiguous(), 2)
        ctx.save_for_backward(x2, w, h)
        ctx.x_shape = x.shape
        return y.reshape(*x.shape[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        x2, w, h = ctx.saved_tensors
        ext = get_ext(required=True)
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        # h is POST-bias: reuse the fused gelu-bwd(+colsum) with a zero
        # bias vector
        zb = torch.zeros(w.shape[0], dtype=h.dtype, device=h.device)
        dh, db = ext.bias_gelu_bwd_db(dy2, h, zb)
        dh = dh.contiguous()
        dx = _hand_dgrad(dh, w).reshape(ctx.x_shape)
        dw = dh.t() @ x2
        return dx, dw, db.to(dy.dtype)


class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        ctx.has_bias = b is not None
        return F.linear(x, w, b)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        dx = (dy2 @ w).reshape(x.shape)
        # wgrad GEMM form picked by shape (tools/bench_linear_bwd.py):
        # wide-K projections (mlp_out 3072->768) run 11% faster as
        # mm(x^T, dy)^T (hipBLASLt sees the other TN problem); the rest
        # as the direct dy^T @ x.
        if x2.shape[1] > dy2.shape[1]:
            dw = torch.mm(x2.t(), dy2.contiguous()).t().contiguous()
        else:
            dw = dy2.t() @ x2
        db = None
        if ctx.has_bias:
            ext = get_ext(required=False)
            if dy.is_cuda and dy.dtype == torch.bfloat16 and ext is not None:
                db = ext.colsum_bf16(dy2.contiguous()).to(dy.dtype)
            else:
                db = dy2.sum(0)
        return dx, dw, db


class Linear(nn.Module):
    """Drop-in nn.Linear replacement (same state-dict keys)."""
    _is_leaf_module = True

    def __init__(self, in_features: int, out_features: int, bias: bool = True):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, x):
        if x.is_cuda:
            if hand_gemm_enabled() and _hand_ok(x, self.weight):
                return _HandLinearFn.apply(x, self.weight, self.bias)
            return _LinearFn.apply(x, self.weight, self.bias)
        return F.linear(x, self.weight, self.bias)


def make_linear(in_features: int, out_features: int, bias: bool = True):
    """Projection factory for the model zoo: nn.Linear by default,
    ops.Linear under RAVNEST_EXPLICIT_LINEAR=1 or the hand-GEMM flag
    (same state-dict keys either way).

    Measured both ways twice: the explicit backward wins ISOLATED per-op
    timings on every BERT-base shape (profiles/r02_linear_bwd.txt — qkv
    0.81 vs 0.84 ms, mlp_out 0.81 vs 0.93) but LOSES end-to-end inside
    the captured training step (BERT 1552 vs 1637 samples/s, GPT-2 560
    vs 610 — hipBLASLt/TunableOp pick different algorithms in the graph
    context), so autograd's addmm backward stays the default."""
    if hand_gemm_enabled() or \
            os.environ.get("RAVNEST_EXPLICIT_LINEAR", "0") == "1":
        return Linear(in_features, out_features, bias=bias)
    return nn.Linear(in_features, out_features, bias=bias)
