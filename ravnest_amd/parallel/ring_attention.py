"""Ring attention: exact attention over a sequence sharded across ranks.

Each rank holds the (B, H, S_local, D) shards of q, k, v for its slice
of the sequence. KV chunks travel around a ring (one P2P shift per
step); every rank computes a partial flash attention of its LOCAL
queries against the resident chunk and merges the partials with the
standard log-sum-exp algebra — mathematically exact, never materializes
S_global x S_global scores, and the per-chunk compute is the EXISTING
fused kernel (csrc/attention.hip returns the per-row lse; the backward
kernel accepts a caller-provided GLOBAL lse, csrc/attention_bwd.hip, so
the chunked backward recomputes the true global probabilities
P = exp(s - lse) chunk by chunk).

Backward: a second ring pass. dq accumulates locally; each traveling
chunk carries its (dk, dv) accumulators and arrives home after a full
revolution (the standard ring-attention gradient flow).

The reference has no sequence/context parallelism (SURVEY.md section
2.2); this adds the axis on top of the comm design as section 5
anticipated. Works over RCCL (xGMI P2P) on GPUs and gloo on CPU; on a
gloo wire CUDA tensors hop through host staging (same rule as
comm/p2p.py Channel.wire_cpu).
"""
from __future__ import annotations

import math

import torch
import torch.distributed as dist
import torch.nn as nn

from ..ops._ext import get_ext


def merge_partials(o_a, lse_a, o_b, lse_b):
    """Merge two normalized attention partials (o, lse) -> (o, lse).

    o_x = softmax_chunk(s_x) @ V_x and lse_x = logsumexp(s_x), so the
    true numerator is o_x * exp(lse_x); the merge renormalizes in a
    shifted basis. Rows where both sides are fully masked (-inf lse)
    come back as zeros with -inf lse.
    """
    m = torch.maximum(lse_a, lse_b)
    m_safe = torch.where(torch.isfinite(m), m, torch.zeros_like(m))
    wa = torch.exp(lse_a - m_safe)
    wb = torch.exp(lse_b - m_safe)
    den = (wa + wb).clamp_min(1e-38)
    o = (o_a * wa.unsqueeze(-1) + o_b * wb.unsqueeze(-1)) / den.unsqueeze(-1)
    lse = m_safe + torch.log(den)
    lse = torch.where(torch.isfinite(m), lse,
                      torch.full_like(lse, float("-inf")))
    return o, lse


def _ring_shift(tensors, group, dst, src):
    """Send `tensors` to group-rank dst, receive from group-rank src."""
    backend = dist.get_backend(group)
    # P2POp peers are GLOBAL ranks
    dst = dist.get_global_rank(group, dst)
    src = dist.get_global_rank(group, src)
    outs = []
    ops = []
    bufs = []
    for t in tensors:
        send = t.contiguous()
        if t.is_cuda and backend != "nccl":
            send = send.cpu()  # gloo wire: host staging
        buf = torch.empty_like(send)
        ops.append(dist.P2POp(dist.isend, send, dst, group=group))
        ops.append(dist.P2POp(dist.irecv, buf, src, group=group))
        bufs.append(buf)
    for req in dist.batch_isend_irecv(ops):
        req.wait()
    for t, buf in zip(tensors, bufs):
        outs.append(buf.to(t.device) if buf.device != t.device else buf)
    return outs


def _use_kernel(q):
    return (q.is_cuda and q.dtype == torch.bfloat16 and q.shape[-1] == 64
            and get_ext(required=False) is not None)


def _chunk_fwd(q, k, v, mask_cols, diag_causal, scale):
    """Partial attention of local q against ONE kv chunk -> (o, lse)."""
    if _use_kernel(q):
        ext = get_ext(required=True)
        o, lse = ext.attn_fwd(
            q, k.contiguous(), v.contiguous(),
            mask_cols if mask_cols is not None else torch.Tensor(),
            diag_causal, scale)
        return o, lse
    s = (q.float() @ k.float().transpose(-2, -1)) * scale
    if diag_causal:
        S = q.shape[-2]
        cm = torch.triu(torch.full((S, S), float("-inf"),
                                   device=q.device), diagonal=1)
        s = s + cm
    if mask_cols is not None:
        s = s + mask_cols.float()
    lse = torch.logsumexp(s, dim=-1)
    # fully-masked rows: lse = -inf; shifting by 0 keeps exp(-inf) = 0
    lse_safe = torch.where(torch.isfinite(lse), lse, torch.zeros_like(lse))
    p = torch.exp(s - lse_safe.unsqueeze(-1))
    o = (p @ v.float()).to(q.dtype)
    return o, lse


def _chunk_bwd(q, k, v, o, do, lse, mask_cols, diag_causal, scale):
    """(dq, dk, dv) of one chunk using the GLOBAL lse."""
    if _use_kernel(q):
        ext = get_ext(required=True)
        dq, dk, dv = ext.attn_bwd(
            q, k.contiguous(), v.contiguous(), o, do, lse,
            mask_cols if mask_cols is not None else torch.Tensor(),
            diag_causal, scale)
        return dq, dk, dv
    s = (q.float() @ k.float().transpose(-2, -1)) * scale
    if diag_causal:
        S = q.shape[-2]
        cm = torch.triu(torch.full((S, S), float("-inf"),
                                   device=q.device), diagonal=1)
        s = s + cm
    if mask_cols is not None:
        s = s + mask_cols.float()
    lse_safe = torch.where(torch.isfinite(lse), lse, torch.zeros_like(lse))
    p = torch.exp(s - lse_safe.unsqueeze(-1))
    dof = do.float()
    dv = (p.transpose(-2, -1) @ dof).to(v.dtype)
    dp = dof @ v.float().transpose(-2, -1)
    delta = (dof * o.float()).sum(-1, keepdim=True)
    ds = p * (dp - delta) * scale
    dq = (ds @ k.float()).to(q.dtype)
    dk = (ds.transpose(-2, -1) @ q.float()).to(k.dtype)
    return dq, dk, dv


def _mask_slice(mask, world, chunk_idx):
    if mask is None:
        return None
    Sg = mask.shape[-1]
    Sc = Sg // world
    return mask[..., chunk_idx * Sc:(chunk_idx + 1) * Sc].contiguous()


class _RingAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, mask, causal, scale, group):
        world = dist.get_world_size(group)
        rank = dist.get_rank(group)
        nxt = (rank + 1) % world
        prv = (rank - 1) % world
        k_c, v_c = k.contiguous(), v.contiguous()
        acc_o = acc_lse = None
        for step in range(world):
            chunk = (rank - step) % world
            skip = causal and chunk > rank
            if not skip:
                o_i, lse_i = _chunk_fwd(
                    q, k_c, v_c, _mask_slice(mask, world, chunk),
                    causal and chunk == rank, scale)
                if acc_o is None:
                    acc_o, acc_lse = o_i, lse_i
                else:
                    acc_o, acc_lse = merge_partials(acc_o, acc_lse,
                                                    o_i, lse_i)
            if step + 1 < world:
                k_c, v_c = _ring_shift([k_c, v_c], group, nxt, prv)
        # merges promote to fp32; the backward kernel reads O in the
        # input dtype
        acc_o = acc_o.to(q.dtype).contiguous()
        ctx.save_for_backward(q, k, v, acc_o, acc_lse,
                              mask if mask is not None else None)
        ctx.causal = causal
        ctx.scale = scale
        ctx.group = group
        return acc_o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse, mask = ctx.saved_tensors
        causal, scale, group = ctx.causal, ctx.scale, ctx.group
        world = dist.get_world_size(group)
        rank = dist.get_rank(group)
        nxt = (rank + 1) % world
        prv = (rank - 1) % world
        do = do.contiguous()
        k_c, v_c = k.contiguous(), v.contiguous()
        dk_c = torch.zeros_like(k_c)
        dv_c = torch.zeros_like(v_c)
        dq = torch.zeros_like(q)
        for step in range(world):
            chunk = (rank - step) % world
            skip = causal and chunk > rank
            if not skip:
                dq_i, dk_i, dv_i = _chunk_bwd(
                    q, k_c, v_c, o, do, lse,
                    _mask_slice(mask, world, chunk),
                    causal and chunk == rank, scale)
                dq += dq_i
                dk_c += dk_i
                dv_c += dv_i
            # one shift per step: after `world` shifts every chunk's
            # accumulated (dk, dv) is back at its owner (world 1: the
            # chunk is already home; a send-to-self would deadlock)
            if world > 1:
                k_c, v_c, dk_c, dv_c = _ring_shift(
                    [k_c, v_c, dk_c, dv_c], group, nxt, prv)
        return dq, dk_c, dv_c, None, None, None, None


def ring_attention(q, k, v, mask=None, causal=False, scale=None,
                   group=None):
    """Exact attention over a sequence sharded across the ranks of
    `group` (default: WORLD). q, k, v: this rank's (B, H, S_local, D)
    shards, equal S_local per rank; mask: additive, broadcastable to
    (B, 1, 1, S_global), replicated on every rank. Returns the local
    (B, H, S_local, D) output slice; gradients flow to the local
    shards."""
    if group is None:
        group = dist.group.WORLD
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    return _RingAttnFn.apply(q.contiguous(), k.contiguous(),
                             v.contiguous(), mask, causal, scale, group)


class RingAttention(nn.Module):
    """Module wrapper over ring_attention (sequence-parallel axis)."""
    _is_leaf_module = True

    def __init__(self, causal: bool = False, group=None):
        super().__init__()
        self.causal = causal
        self.group = group

    def forward(self, q, k, v, mask=None):
        return ring_attention(q, k, v, mask=mask, causal=self.causal,
                              group=self.group)
