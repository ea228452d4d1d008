"""Fused embedding stems: gather(word) + gather(pos) [+ LayerNorm].

One HBM pass replaces torch's gather/gather/add/LN chain, and the
backward RE-GATHERS the pre-LN sum instead of keeping the (B,S,H)
activation alive (csrc/embedding.hip). dword is a scatter-add (fp32
workspace atomics), dpos a no-atomic batch reduction.

Workload parity: reference BERT/minGPT embedding stems — SURVEY.md
section 2.3 "Embedding + positional lookup" row.
"""
from __future__ import annotations

import torch

from ._ext import get_ext


class _EmbLNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, ids, word, pos, w, b, eps):
        ext = get_ext(required=True)
        y, mean, rstd = ext.emb2_ln_fwd(ids, word, pos, w, b, eps)
        ctx.save_for_backward(ids, word, pos, w, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        ids, word, pos, w, mean, rstd = ctx.saved_tensors
        ext = get_ext(required=True)
        # recompute the pre-LN sum (2-row gather) instead of storing it
        x = ext.emb2_add_fwd(ids, word, pos)
        dx, dw, db = ext.layernorm_bwd(dy.contiguous(), x, w, mean, rstd)
        dword, dpos = ext.emb2_bwd(dx, ids, word.size(0), pos.size(0))
        return (None, dword.to(word.dtype), dpos.to(pos.dtype),
                dw, db, None)


class _EmbAddFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, ids, word, pos):
        ext = get_ext(required=True)
        ctx.save_for_backward(ids)
        ctx.vp = (word.size(0), pos.size(0))
        ctx.dtypes = (word.dtype, pos.dtype)
        return ext.emb2_add_fwd(ids, word, pos)

    @staticmethod
    def backward(ctx, dy):
        (ids,) = ctx.saved_tensors
        ext = get_ext(required=True)
        dword, dpos = ext.emb2_bwd(dy.contiguous(), ids, *ctx.vp)
        return None, dword.to(ctx.dtypes[0]), dpos.to(ctx.dtypes[1])


def _fusable(ids, word):
    return (ids.is_cuda and word.dtype == torch.bfloat16
            and word.size(1) % 8 == 0 and word.size(1) <= 1024)


def embedding_ln(ids, word, pos, weight, bias, eps):
    """LN(word[ids] + pos[:S]) — the BERT stem."""
    if _fusable(ids, word):
        # full pos TABLE goes in (kernel indexes row s = token position);
        # dpos then matches the parameter shape
        return _EmbLNFn.apply(ids.contiguous(), word, pos, weight, bias,
                              eps)
    from .layernorm import layer_norm
    S = ids.size(1)
    x = torch.nn.functional.embedding(ids, word) + pos[:S]
    return layer_norm(x, weight, bias, eps)


def embedding_add(ids, word, pos):
    """word[ids] + pos[:S] — the GPT stem (LN lives in the blocks)."""
    if _fusable(ids, word):
        return _EmbAddFn.apply(ids.contiguous(), word, pos)
    S = ids.size(1)
    return torch.nn.functional.embedding(ids, word) + pos[:S]
