"""Sequence-parallel GPT: train a causal LM with the SEQUENCE sharded
across ranks.

Everything in a transformer block except attention is token-local
(LayerNorm, projections, GELU, embeddings, the LM head), so sequence
parallelism only needs (1) ring attention in the core, (2) absolute
position offsets in the embeddings, and (3) a gradient all-reduce over
the SP group after backward (each rank's tokens contribute a partial
parameter gradient). `sequence_parallelize()` rewrites a stock GPT
in place for a given (rank, world) shard; existing single-process
paths are untouched.

The reference has no SP axis; this builds it on
ring_attention (same package) and the unchanged model/op stack.
"""
from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn

from .ring_attention import ring_attention


class RingCoreQKV(nn.Module):
    """Drop-in replacement for ops.AttentionCoreQKV over a sequence
    shard: unpacks (B, S_local, 3, H, D) qkv and runs ring attention
    across the SP group; returns token-major (B, S_local, H*D)."""
    _is_leaf_module = True

    def __init__(self, causal: bool = True, group=None):
        super().__init__()
        self.causal = causal
        self.group = group

    def forward(self, qkv, mask=None):
        q, k, v = qkv.unbind(dim=2)          # (B, S_local, H, D)
        q = q.transpose(1, 2).contiguous()   # (B, H, S_local, D)
        k = k.transpose(1, 2).contiguous()
        v = v.transpose(1, 2).contiguous()
        o = ring_attention(q, k, v, mask=mask, causal=self.causal,
                           group=self.group)
        return o.transpose(1, 2).flatten(2)


def sequence_parallelize(model, group=None):
    """Rewrite a models.gpt.GPT IN PLACE for sequence-parallel training:
    every block's attention core becomes ring attention over `group`,
    and the position table is offset so this rank's tokens see their
    ABSOLUTE positions (forward then takes the local (B, S_local) shard
    of the token ids). Returns the model.

    Note: the fused GPU embedding stem indexes positions from 0, so SP
    models route through the eager embedding path (position offsets via
    the pos_ids buffer)."""
    if group is None:
        group = dist.group.WORLD
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    cfg = model.cfg
    if cfg.block_size % world:
        raise ValueError("block_size must divide by the SP world size")
    s_local = cfg.block_size // world
    off = rank * s_local
    emb = model.embeddings
    with torch.no_grad():
        emb.pos_ids = torch.arange(
            off, off + s_local, device=emb.pos_ids.device).unsqueeze(0)
    # fused GPU stem has no offset support: force the eager gather path
    emb._sp_offset = off
    for blk in model.blocks:
        blk.core = RingCoreQKV(causal=True, group=group)
    model._sp_group = group
    model._sp_local = s_local
    return model


def allreduce_gradients(model, group=None):
    """Sum parameter gradients over the SP group (each rank's sequence
    shard contributes a partial gradient to the SHARED weights). Call
    between backward() and optimizer.step()."""
    if group is None:
        group = getattr(model, "_sp_group", dist.group.WORLD)
    for p in model.parameters():
        if p.grad is not None:
            dist.all_reduce(p.grad, group=group)
