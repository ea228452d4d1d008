"""Convolution helpers.

A 1x1 stride-1 convolution IS a GEMM over (C_in -> C_out) per pixel —
MIOpen still runs it through per-image im2col + GEMM (profile r01:
thousands of Im2d2Col launches per ResNet step). Conv1x1 routes it
straight through torch.matmul (hipBLASLt), state-dict compatible with
nn.Conv2d(k=1). Spatial convolutions stay on MIOpen (the library path;
hand-written implicit-GEMM conv kernels are round-2 work).
"""
from __future__ import annotations

import torch
import torch.nn as nn


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
