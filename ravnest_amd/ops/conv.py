"""Convolution helpers.

A 1x1 stride-1 convolution IS a GEMM over (C_in -> C_out) per pixel —
MIOpen still runs it through per-image im2col + GEMM (profile r01:
thousands of Im2d2Col launches per ResNet step). Conv1x1 routes it
straight through torch.matmul (hipBLASLt), state-dict compatible with
nn.Conv2d(k=1). Spatial convolutions: fp32 stays on MIOpen (measured
faster); bf16 has the hand-written implicit-GEMM MFMA path
(conv2d_mfma, csrc/conv.hip).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ._ext import get_ext


class Conv1x1(nn.Module):
    """nn.Conv2d(in, out, 1, stride=s, bias=False) drop-in (NCHW)."""
    _is_leaf_module = True

    def __init__(self, in_ch: int, out_ch: int, stride: int = 1):
        super().__init__()
        self.in_channels = in_ch
        self.out_channels = out_ch
        self.stride = stride
        self.weight = nn.Parameter(torch.empty(out_ch, in_ch, 1, 1))
        nn.init.kaiming_normal_(self.weight, mode="fan_out",
                                nonlinearity="relu")

    def forward(self, x):
        if self.stride != 1:
            x = x[:, :, ::self.stride, ::self.stride]
        N, C, H, W = x.shape
        w = self.weight.view(self.out_channels, C)
        out = torch.matmul(w, x.reshape(N, C, H * W))
        return out.reshape(N, self.out_channels, H, W)


class _ConvMFMAFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, stride, padding):
        ext = get_ext(required=True)
        y = ext.conv2d_fwd(x, weight, bias, stride, padding, False)
        ctx.save_for_backward(x, weight)
        ctx.meta = (stride, padding, bias is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        stride, padding, has_bias = ctx.meta
        ext = get_ext(required=True)
        dy = dy.contiguous()
        dx = ext.conv2d_dgrad(dy, weight, x.size(0), x.size(2), x.size(3),
                              stride, padding)
        dw = ext.conv2d_wgrad(dy, x, weight.size(2), weight.size(3),
                              stride, padding).to(weight.dtype)
        db = dy.float().sum(dim=(0, 2, 3)) if has_bias else None
        return dx, dw, db, None, None


def conv2d_mfma(x, weight, bias=None, stride=1, padding=0):
    """Hand-written implicit-GEMM MFMA convolution (bf16 NCHW,
    csrc/conv.hip — SURVEY.md section 2.3 conv row). Component path: the
    default ResNet pipeline stays fp32 MIOpen (measured faster at fp32);
    this targets bf16 convs, where MIOpen measured ~4x slower than fp32
    on this stack."""
    if x.is_cuda and x.dtype == torch.bfloat16:
        b = bias.float() if bias is not None else None
        return _ConvMFMAFn.apply(x.contiguous(), weight.contiguous(), b,
                                 stride, padding)
    return torch.nn.functional.conv2d(x, weight, bias, stride, padding)
