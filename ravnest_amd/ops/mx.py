"""MX-scaled fp8 (OCP e4m3) linear layer on the gfx950 block-scaled MFMA.

Forward quantizes activations and weights to e4m3 with per-32-element
e8m0 block scales and runs `mfma_scale_f32_32x32x64_f8f6f4` (2x the bf16
MFMA rate, csrc/mx.hip). Backward stays bf16 (dX = dY @ W, dW = dY^T @ X
— standard fp8-training recipe: low-precision forward, higher-precision
gradients). CPU / non-multiple-of-64 shapes fall back to a plain matmul.

Workload parity: BASELINE.json config "GPT-Sorter fp8 CDNA4 MFMA path".
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ._ext import get_ext


class _MXLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ext = get_ext(required=True)
        shp = x.shape
        x2 = x.reshape(-1, shp[-1]).contiguous()
        xq, xs = ext.mx_quant(x2)
        wq, ws = ext.mx_quant(weight.contiguous())
        K = x2.shape[1]
        if K % 128 == 0 and K >= 256 and xq.numel() < 2**32 \
                and wq.numel() < 2**32:
            # LDS-staged 4-phase kernel (csrc/mx_gemm2.hip): 845-1294 TF
            # vs the register-tiled mx_gemm's ~530
            y = ext.mx_gemm2(xq, xs, wq, ws)
        else:
            y = ext.mx_gemm(xq, xs, wq, ws)
        if bias is not None:
            y = y + bias
        ctx.save_for_backward(x2, weight)
        ctx.has_bias = bias is not None
        return y.reshape(*shp[:-1], weight.size(0))

    @staticmethod
    def backward(ctx, dy):
        x2, weight = ctx.saved_tensors
        d2 = dy.reshape(-1, dy.shape[-1])
        dx = (d2 @ weight).reshape(*dy.shape[:-1], weight.size(1))
        dw = d2.t() @ x2
        db = d2.sum(0) if ctx.has_bias else None
        return dx, dw, db


def mx_linear(x, weight, bias=None):
    if x.is_cuda and x.dtype == torch.bfloat16 and \
            x.size(-1) % 64 == 0 and x.size(-1) == weight.size(1):
        return _MXLinearFn.apply(x, weight, bias)
    return torch.nn.functional.linear(x, weight, bias)


class MXLinear(nn.Module):
    """Drop-in nn.Linear running the MX fp8 forward path on GPU.

    State-dict compatible with nn.Linear (weight/bias stay bf16 masters;
    quantization happens on the fly per forward).
    """

    _is_leaf_module = True  # fx: device-dependent dispatch

    def __init__(self, in_features: int, out_features: int, bias: bool = True):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, x):
        return mx_linear(x, self.weight, self.bias)

    @classmethod
    def from_linear(cls, lin: nn.Linear) -> "MXLinear":
        m = cls(lin.in_features, lin.out_features, lin.bias is not None)
        with torch.no_grad():
            m.weight.copy_(lin.weight)
            if lin.bias is not None:
                m.bias.copy_(lin.bias)
        return m
