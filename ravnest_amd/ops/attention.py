"""Fused multi-head attention core (flash-style forward on CDNA4).

Reference workload parity: minGPT causal self-attention
(model_without_padding_mask.py:50-113) and BERT's padded-mask attention —
SURVEY.md section 2.3. Forward is the hand-written HIP kernel
(csrc/attention.hip): per Q-tile online-softmax over K/V tiles, bf16 MFMA
(32x32x16), XOR-swizzled K LDS tiles, no S x S score matrix in HBM; it
returns O and the log-sum-exp rows. Backward on the D=64 path runs the
fused flash backward kernels (csrc/attention_bwd.hip: delta + dv + dk +
dq, P recomputed from (Q,K,lse) on-chip); other head dims recompute P
with plain hipBLASLt GEMMs + elementwise torch.

Shapes: q,k,v (B, H, S, D); additive mask broadcastable to (B, 1, S, S)
or None; causal flag for GPT-style models.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn

from ._ext import get_ext


def _math_attention(q, k, v, mask, causal, scale):
    s = (q @ k.transpose(-2, -1)) * scale
    if causal:
        S = q.shape[-2]
        cm = torch.full((S, S), float("-inf"), device=q.device)
        cm = torch.triu(cm, diagonal=1)
        s = s + cm
    if mask is not None:
        s = s + mask
    p = torch.softmax(s.float(), dim=-1).to(q.dtype)
    return p @ v


class _AttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, mask, causal, scale):
        ext = get_ext(required=True)
        o, lse = ext.attn_fwd(q, k, v,
                              mask if mask is not None else torch.Tensor(),
                              causal, scale)
        ctx.save_for_backward(q, k, v, mask if mask is not None else None,
                              o, lse)
        ctx.causal = causal
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, mask, o, lse = ctx.saved_tensors
        causal, scale = ctx.causal, ctx.scale
        do = do.contiguous()
        if q.shape[-1] == 64:
            ext = get_ext(required=True)
            dq, dk, dv = ext.attn_bwd(
                q, k, v, o, do, lse,
                mask if mask is not None else torch.Tensor(), causal, scale)
            return dq, dk, dv, None, None, None
        # head_dim != 64: recompute P from lse via library GEMMs
        # P = exp(S*scale + mask - lse)
        s = (q @ k.transpose(-2, -1)).float() * scale
        if causal:
            S = q.shape[-2]
            cm = torch.triu(torch.full((S, S), float("-inf"),
                                       device=q.device), diagonal=1)
            s = s + cm
        if mask is not None:
            s = s + mask.float()
        p = torch.exp(s - lse.unsqueeze(-1)).to(q.dtype)
        dv = p.transpose(-2, -1) @ do
        dp = (do @ v.transpose(-2, -1)).float()
        delta = (do.float() * o.float()).sum(-1, keepdim=True)
        ds = (p.float() * (dp - delta)) * scale
        ds = ds.to(q.dtype)
        dq = ds @ k
        dk = ds.transpose(-2, -1) @ q
        return dq, dk, dv, None, None, None


def attention(q, k, v, mask=None, causal=False, scale=None):
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        return _AttnFn.apply(q.contiguous(), k.contiguous(), v.contiguous(),
                             mask, causal, scale)
    return _math_attention(q, k, v, mask, causal, scale)


class AttentionCore(nn.Module):
    _is_leaf_module = True

    def __init__(self, causal: bool = False):
        super().__init__()
        self.causal = causal

    def forward(self, q, k, v, mask=None):
        return attention(q, k, v, mask=mask, causal=self.causal)


class _AttnQKVFn(torch.autograd.Function):
    """Packed path: qkv (B,S,3,H,D) -> o (B,S,H,D). No permute/contiguous
    copies: the kernel reads the projection's natural layout and writes O
    in token-major order (the next Linear's input layout).

    pdrop > 0 fuses attention-PROB dropout (HF BertSelfAttention
    semantics) into the fwd and bwd kernels: the philox seed is drawn
    from the torch CPU generator (so the versioned-recompute engine's
    fork_rng replay redraws the identical mask) and XORed with the
    device graph counter at run time (fresh masks per hipGraph replay).
    The fwd publishes its keep-mask as one 32-key word per (bh, qrow)
    k-block; the three bwd kernels READ those words (regenerating via
    philox measured ~4x the masking cost)."""

    @staticmethod
    def forward(ctx, qkv, mask, causal, scale, pdrop=0.0):
        ext = get_ext(required=True)
        seed = 0
        seed_buf = None
        if pdrop > 0:
            from . import rng
            seed = int(torch.randint(0, 2**62, (1,)).item())
            seed_buf = rng.device_seed_counter(qkv.device)
        res = ext.attn_fwd_qkv(qkv,
                               mask if mask is not None else torch.Tensor(),
                               causal, scale, pdrop, seed, seed_buf)
        o, lse = res[0], res[1]
        mbits = res[2] if len(res) > 2 else None
        ctx.save_for_backward(qkv, mask if mask is not None else None, o,
                              lse, mbits)
        ctx.causal = causal
        ctx.scale = scale
        ctx.pdrop = pdrop
        return o

    @staticmethod
    def backward(ctx, do):
        qkv, mask, o, lse, mbits = ctx.saved_tensors
        ext = get_ext(required=True)
        dqkv = ext.attn_bwd_qkv(
            qkv, o, do.contiguous(), lse,
            mask if mask is not None else torch.Tensor(),
            ctx.causal, ctx.scale, ctx.pdrop, mbits)
        return dqkv, None, None, None, None


def attention_qkv(qkv, mask=None, causal=False, scale=None,
                  prob_dropout=0.0):
    """qkv (B,S,3,H,D) -> (B,S,H*D). prob_dropout fuses attention-prob
    dropout into the kernels (D=64 path)."""
    H, D = qkv.shape[-2], qkv.shape[-1]
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    if qkv.is_cuda and D == 64:
        o = _AttnQKVFn.apply(qkv.contiguous(), mask, causal, scale,
                             prob_dropout)
        return o.flatten(2)
    # fallback: unpack + (custom or math) attention
    q, k, v = (qkv.permute(2, 0, 3, 1, 4)[i] for i in range(3))
    o = attention(q.contiguous(), k.contiguous(), v.contiguous(),
                  mask=mask, causal=causal, scale=scale)
    return o.transpose(1, 2).flatten(2)


def attention_qkv_prob_dropout(qkv, mask, causal, scale, p, training):
    """Attention with dropout ON THE PROBABILITIES — exact HuggingFace
    BertSelfAttention semantics. Composed path: the S x S probs are
    materialized and masked by the replayable philox dropout kernel
    (csrc/dropout.hip), so the versioned-recompute engine replays the
    identical mask. Slower than the fused flash kernel (which cannot
    drop individual probs without a P-materialization); use when exact
    prob-dropout semantics matter more than throughput."""
    from .dropout import dropout as _phil_dropout
    q, k, v = qkv.unbind(dim=2)          # (B, S, H, D) each
    q = q.transpose(1, 2)                # (B, H, S, D)
    k = k.transpose(1, 2)
    v = v.transpose(1, 2)
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    s = (q @ k.transpose(-2, -1)) * scale
    if causal:
        S = q.shape[-2]
        cm = torch.triu(torch.full((S, S), float("-inf"),
                                   device=q.device), diagonal=1)
        s = s + cm
    if mask is not None:
        s = s + mask
    probs = torch.softmax(s.float(), dim=-1).to(qkv.dtype)
    if training and p > 0:
        probs = _phil_dropout(probs, p, training=True)
    o = probs @ v                        # (B, H, S, D)
    # token-major (B, S, H*D) like the fused kernel (the next Linear's
    # input layout)
    B, Hh, S, D = o.shape
    return o.transpose(1, 2).reshape(B, S, Hh * D)


class AttentionCoreQKV(nn.Module):
    """Fused flash-style attention over packed (B,S,3,H,D) qkv.

    prob_dropout: dropout rate applied to the ATTENTION PROBABILITIES
    (HF BertSelfAttention semantics). On the D=64 path the philox mask
    is FUSED into the flash fwd/bwd kernels (no P materialization;
    masks regenerated in the backward from the same counter scheme);
    other shapes route through the composed P-materializing path.
    Measured cost on BERT-base: ~19% step time (philox at 4 kernel
    sites), so prob_dropout == 0 remains the perf default used by the
    bench models, which apply dropout AFTER the output projection — a
    DOCUMENTED semantics difference vs HF BERT."""
    _is_leaf_module = True

    def __init__(self, causal: bool = False, prob_dropout: float = 0.0):
        super().__init__()
        self.causal = causal
        self.prob_dropout = prob_dropout

    def forward(self, qkv, mask=None):
        if self.prob_dropout > 0 and self.training and qkv.is_cuda:
            if qkv.shape[-1] == 64:  # fused philox masks in the kernels
                return attention_qkv(qkv, mask=mask, causal=self.causal,
                                     prob_dropout=self.prob_dropout)
            return attention_qkv_prob_dropout(
                qkv, mask, self.causal, None, self.prob_dropout,
                self.training)
        return attention_qkv(qkv, mask=mask, causal=self.causal)
