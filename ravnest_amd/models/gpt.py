"""GPT decoder family (gpt-nano ... gpt2-small) on the MI355X op library.

Workload parity: the reference's minGPT sorter example
(examples/sorter/mingpt/model_without_padding_mask.py) including the
fx-friendly structure (custom-op modules are fx leaves — the reference
patched arange/masked_fill into leaf modules for the same reason,
model_without_padding_mask.py:34-48). Causal attention runs on the fused
CDNA4 attention kernel.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import (AttentionCoreQKV, Dropout, FusedLayerNorm,
                   LinearGelu, MXLinear)
from ..ops import embedding_add


class GPTConfig:
    def __init__(self, vocab_size, block_size, n_layer, n_head, n_embd,
                 dropout=0.1, fp8=False):
        self.vocab_size = vocab_size
        self.block_size = block_size
        self.n_layer = n_layer
        self.n_head = n_head
        self.n_embd = n_embd
        self.dropout = dropout
        # fp8=True runs the block projections on the MX-scaled fp8 MFMA
        # path (ops.MXLinear, state-dict compatible with nn.Linear) —
        # BASELINE.json "GPT-Sorter fp8 CDNA4 MFMA path". Requires
        # n_embd % 64 == 0.
        self.fp8 = fp8

    @classmethod
    def nano(cls, vocab_size=16, block_size=16):
        return cls(vocab_size, block_size, n_layer=3, n_head=3, n_embd=48)

    @classmethod
    def nano64(cls, vocab_size=16, block_size=16):
        """gpt-nano scale with head_dim 64 (the GPU attention kernel's
        native size)."""
        return cls(vocab_size, block_size, n_layer=3, n_head=3, n_embd=192)

    @classmethod
    def gpt2_small(cls, vocab_size=50257, block_size=1024):
        return cls(vocab_size, block_size, n_layer=12, n_head=12, n_embd=768)


class GPTEmbeddings(nn.Module):
    _is_leaf_module = True  # fx: data-dependent position slice

    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.tok_emb = nn.Embedding(cfg.vocab_size, cfg.n_embd)
        self.pos_emb = nn.Embedding(cfg.block_size, cfg.n_embd)
        self.drop = Dropout(cfg.dropout)
        self.register_buffer("pos_ids",
                             torch.arange(cfg.block_size).unsqueeze(0),
                             persistent=False)

    def forward(self, idx):
        # sequence-parallel shards carry a position offset in pos_ids
        # (parallel/sp.py); the fused stem indexes positions from 0, so
        # SP routes through the eager gather
        if idx.is_cuda and self.tok_emb.weight.dtype == torch.bfloat16 \
                and not getattr(self, "_sp_offset", 0):
            x = embedding_add(idx, self.tok_emb.weight, self.pos_emb.weight)
            return self.drop(x)
        S = idx.size(1)
        x = self.tok_emb(idx) + self.pos_emb(self.pos_ids[:, :S])
        return self.drop(x)


class GPTBlock(nn.Module):
    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.n_head = cfg.n_head
        self.head_dim = cfg.n_embd // cfg.n_head
        fp8 = getattr(cfg, "fp8", False)
        if fp8 and cfg.n_embd % 64 != 0:
            raise ValueError("fp8 GPT needs n_embd % 64 == 0")
        from ..ops.linear import make_linear
        lin = MXLinear if fp8 else make_linear
        self.ln1 = FusedLayerNorm(cfg.n_embd)
        self.qkv = lin(cfg.n_embd, 3 * cfg.n_embd)
        self.core = AttentionCoreQKV(causal=True)
        self.proj = lin(cfg.n_embd, cfg.n_embd)
        self.attn_drop = Dropout(cfg.dropout)
        self.ln2 = FusedLayerNorm(cfg.n_embd)
        self.mlp_in = LinearGelu(cfg.n_embd, 4 * cfg.n_embd)
        self.mlp_out = lin(4 * cfg.n_embd, cfg.n_embd)
        self.mlp_drop = Dropout(cfg.dropout)

    def forward(self, x):
        h = self.ln1(x)
        qkv = self.qkv(h).unflatten(-1, (3, self.n_head, self.head_dim))
        o = self.core(qkv)
        x = x + self.attn_drop(self.proj(o))
        x = x + self.mlp_drop(self.mlp_out(self.mlp_in(self.ln2(x))))
        return x


class GPT(nn.Module):
    """idx (B,S) int64 -> logits (B,S,V)."""

    def __init__(self, cfg: GPTConfig):
        super().__init__()
        self.cfg = cfg
        self.embeddings = GPTEmbeddings(cfg)
        self.blocks = nn.ModuleList(
            [GPTBlock(cfg) for _ in range(cfg.n_layer)])
        self.ln_f = FusedLayerNorm(cfg.n_embd)
        from ..ops.linear import make_linear
        self.lm_head = make_linear(cfg.n_embd, cfg.vocab_size, bias=False)
        self.apply(self._init)

    @staticmethod
    def _init(m):
        from ..ops import Linear as _OpsLinear
        if isinstance(m, (nn.Linear, _OpsLinear)):
            nn.init.normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, std=0.02)

    def forward(self, idx):
        x = self.embeddings(idx)
        for b in self.blocks:
            x = b(x)
        return self.lm_head(self.ln_f(x))

    @torch.no_grad()
    def generate(self, idx, max_new_tokens):
        """Greedy decode (parity: sorter inference,
        examples/sorter/mingpt/utils.py)."""
        self.eval()
        for _ in range(max_new_tokens):
            idx_cond = idx[:, -self.cfg.block_size:]
            logits = self(idx_cond)
            nxt = logits[:, -1, :].argmax(dim=-1, keepdim=True)
            idx = torch.cat([idx, nxt], dim=1)
        return idx
