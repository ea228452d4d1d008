"""Fused BatchNorm2d (NCHW) on CDNA4.

Reference workload parity: BatchNorm in CNN/ResNet/Inception (SURVEY.md
section 2.3 "standalone BN fwd/bwd with Welford"). GPU path:
csrc/batchnorm.hip (per-channel block reductions, fp32 stats); CPU path:
torch native. Drop-in state-dict compatible with nn.BatchNorm2d.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ._ext import get_ext


class _BatchNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, running_mean, running_var, training, momentum,
                eps):
        ext = get_ext(required=True)
        y, mean, rstd = ext.batchnorm_fwd(x, w.float(), b.float(),
                                          running_mean, running_var,
                                          training, momentum, eps)
        ctx.save_for_backward(x, w, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, mean, rstd = ctx.saved_tensors
        ext = get_ext(required=True)
        dx, dw, db = ext.batchnorm_bwd(dy.contiguous(), x, w.float(), mean,
                                       rstd)
        return dx, dw.to(w.dtype), db.to(w.dtype), None, None, None, None, \
            None


class FusedBatchNorm2d(nn.Module):
    _is_leaf_module = True

    def __init__(self, num_features: int, eps: float = 1e-5,
                 momentum: float = 0.1):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))
        self.register_buffer("num_batches_tracked",
                             torch.tensor(0, dtype=torch.long))

    def forward(self, x):
        if x.is_cuda:
            # stat buffers stay fp32 even under model.to(bf16): the kernel
            # updates them in place at fp32 precision
            if self.running_mean.dtype != torch.float32:
                self.running_mean.data = self.running_mean.data.float()
                self.running_var.data = self.running_var.data.float()
            if self.training:
                self.num_batches_tracked += 1
            return _BatchNormFn.apply(x.contiguous(), self.weight, self.bias,
                                      self.running_mean, self.running_var,
                                      self.training, self.momentum, self.eps)
        return F.batch_norm(x, self.running_mean, self.running_var,
                            self.weight, self.bias, self.training,
                            self.momentum, self.eps)
