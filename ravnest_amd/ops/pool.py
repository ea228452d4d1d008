"""Pooling on hand-written CDNA4 kernels (csrc/pool.hip).

MaxPool2d / AvgPool2d (square window) and the adaptive-(1,1) global
average pool the reference workloads use (ResNet stem & head, Inception
branches, CNN — SURVEY.md section 2.3 pooling row). Backwards are
gather-formulated (no atomics, deterministic). CPU falls back to torch.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ._ext import get_ext


class _MaxPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k, s, p):
        ext = get_ext(required=True)
        y, idx = ext.maxpool2d_fwd(x, k, s, p)
        ctx.save_for_backward(idx)
        ctx.meta = (x.shape[2], x.shape[3], k, s, p)
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        h, w, k, s, p = ctx.meta
        ext = get_ext(required=True)
        return (ext.maxpool2d_bwd(dy.contiguous(), idx, h, w, k, s, p),
                None, None, None)


class _AvgPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k, s, p, include_pad):
        ext = get_ext(required=True)
        ctx.meta = (x.shape[2], x.shape[3], k, s, p, include_pad)
        return ext.avgpool2d_fwd(x, k, s, p, include_pad)

    @staticmethod
    def backward(ctx, dy):
        h, w, k, s, p, inc = ctx.meta
        ext = get_ext(required=True)
        return (ext.avgpool2d_bwd(dy.contiguous(), h, w, k, s, p, inc),
                None, None, None, None)


class _GlobalAvgPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = get_ext(required=True)
        ctx.meta = (x.shape[2], x.shape[3])
        return ext.global_avgpool_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        ext = get_ext(required=True)
        return ext.global_avgpool_bwd(dy.contiguous(), *ctx.meta)


class MaxPool2d(nn.Module):
    _is_leaf_module = True  # fx: device-dependent dispatch

    def __init__(self, kernel_size, stride=None, padding=0):
        super().__init__()
        self.k = kernel_size
        self.s = stride if stride is not None else kernel_size
        self.p = padding

    def forward(self, x):
        if x.is_cuda:
            return _MaxPoolFn.apply(x.contiguous(), self.k, self.s, self.p)
        return torch.nn.functional.max_pool2d(x, self.k, self.s, self.p)


class AvgPool2d(nn.Module):
    _is_leaf_module = True  # fx: device-dependent dispatch

    def __init__(self, kernel_size, stride=None, padding=0,
                 count_include_pad=True):
        super().__init__()
        self.k = kernel_size
        self.s = stride if stride is not None else kernel_size
        self.p = padding
        self.inc = count_include_pad

    def forward(self, x):
        if x.is_cuda:
            return _AvgPoolFn.apply(x.contiguous(), self.k, self.s, self.p,
                                    self.inc)
        return torch.nn.functional.avg_pool2d(
            x, self.k, self.s, self.p, count_include_pad=self.inc)


class AdaptiveAvgPool2d(nn.Module):
    """Only the (1,1) target the reference workloads use."""

    _is_leaf_module = True  # fx: device-dependent dispatch

    def __init__(self, output_size=(1, 1)):
        super().__init__()
        if output_size not in ((1, 1), 1):
            raise ValueError("AdaptiveAvgPool2d: only (1,1) supported")

    def forward(self, x):
        if x.is_cuda:
            return _GlobalAvgPoolFn.apply(x.contiguous())
        return torch.nn.functional.adaptive_avg_pool2d(x, (1, 1))
