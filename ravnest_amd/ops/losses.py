"""Fused softmax cross-entropy over large vocabularies.

Reference workload parity: CE with ignore_index for GPT-sorter
(examples/sorter/provider.py:14-15) and BERT's 30522-vocab MLM loss
(examples/bert/provider.py:31-41) — SURVEY.md section 2.3. GPU: one HIP
kernel computes per-row max/logsumexp and the loss (csrc/ce_loss.hip);
backward is the fused (softmax - onehot) * scale kernel, no V-sized
intermediate in fp32.
"""
from __future__ import annotations

import torch

from ._ext import get_ext


class _CrossEntropyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets, ignore_index):
        ext = get_ext(required=True)
        loss_sum, lse, n_valid = ext.ce_fwd(logits, targets, ignore_index)
        ctx.save_for_backward(logits, targets, lse, n_valid)
        ctx.ignore_index = ignore_index
        return loss_sum / n_valid.clamp(min=1)

    @staticmethod
    def backward(ctx, dloss):
        logits, targets, lse, n_valid = ctx.saved_tensors
        ext = get_ext(required=True)
        dlogits = ext.ce_bwd(logits, targets, lse,
                             dloss / n_valid.clamp(min=1).to(dloss.dtype),
                             ctx.ignore_index)
        return dlogits, None, None


def cross_entropy(logits: torch.Tensor, targets: torch.Tensor,
                  ignore_index: int = -100) -> torch.Tensor:
    """Mean CE over non-ignored targets. logits (N, V) any float dtype,
    targets (N,) int64."""
    logits2 = logits.reshape(-1, logits.shape[-1])
    targets2 = targets.reshape(-1)
    if logits.is_cuda:
        return _CrossEntropyFn.apply(logits2.contiguous(),
                                     targets2.contiguous(), ignore_index)
    return torch.nn.functional.cross_entropy(
        logits2.float(), targets2, ignore_index=ignore_index)


class CrossEntropyLoss:
    """Criterion-callable for the leaf stage (Node(criterion=...))."""

    def __init__(self, ignore_index: int = -100):
        self.ignore_index = ignore_index

    def __call__(self, preds, targets):
        if isinstance(targets, (tuple, list)):
            targets = targets[-1]
        return cross_entropy(preds, targets, self.ignore_index)
