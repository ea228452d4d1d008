"""BERT-base encoder for MLM pretraining, built on the MI355X op library.

Workload parity: the reference's BERT pretraining example (HuggingFace
BertForPreTraining on tokenized WikiText, examples/bert/provider.py) —
re-implemented natively so the hot ops (fused LayerNorm, bias-GELU,
flash-style attention, replayable dropout) run on the hand-written CDNA4
kernels while projections use hipBLASLt. fx-traceable for the pipeline
splitter (custom-op modules are fx leaves).

BERT-base config: L=12, H=768, A=12, I=3072, vocab=30522, seq<=512.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import embedding_ln
from ..ops import (AddLayerNorm, AttentionCoreQKV, Dropout, FusedLayerNorm,
                   GELU, LinearGelu)
from ..ops.linear import make_linear


class BertConfig:
    def __init__(self, vocab_size=30522, hidden=768, layers=12, heads=12,
                 intermediate=3072, max_seq=512, type_vocab=2,
                 dropout=0.1, layer_norm_eps=1e-12):
        self.vocab_size = vocab_size
        self.hidden = hidden
        self.layers = layers
        self.heads = heads
        self.intermediate = intermediate
        self.max_seq = max_seq
        self.type_vocab = type_vocab
        self.dropout = dropout
        self.layer_norm_eps = layer_norm_eps

    @classmethod
    def base(cls, **kw):
        return cls(**kw)

    @classmethod
    def tiny(cls, **kw):
        # head_dim 64 (the attention kernel's native size)
        d = dict(vocab_size=1024, hidden=128, layers=2, heads=2,
                 intermediate=256, max_seq=64)
        d.update(kw)
        return cls(**d)


class BertEmbeddings(nn.Module):
    _is_leaf_module = True  # fx: data-dependent position slice

    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.word = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.position = nn.Embedding(cfg.max_seq, cfg.hidden)
        self.ln = FusedLayerNorm(cfg.hidden, cfg.layer_norm_eps)
        self.drop = Dropout(cfg.dropout)
        self.register_buffer(
            "pos_ids", torch.arange(cfg.max_seq).unsqueeze(0),
            persistent=False)

    def forward(self, input_ids):
        if input_ids.is_cuda and self.word.weight.dtype == torch.bfloat16:
            # single-pass gather+gather+LN kernel (ops/embedding.py)
            y = embedding_ln(input_ids, self.word.weight,
                             self.position.weight, self.ln.weight,
                             self.ln.bias, self.ln.eps)
            return self.drop(y)
        S = input_ids.size(1)
        x = self.word(input_ids) + self.position(self.pos_ids[:, :S])
        return self.drop(self.ln(x))


class BertSelfAttention(nn.Module):
    """DOCUMENTED SEMANTICS DIFFERENCE vs HuggingFace BERT: by default
    dropout is applied AFTER the output projection (self.drop below)
    instead of on the attention probabilities — the fused flash kernel
    never materializes the S x S probs. Set RAVNEST_EXACT_ATTN_DROPOUT=1
    (before model construction) for exact HF prob-dropout semantics via
    the composed P-materializing path (replayable philox mask; lower
    throughput — ops/attention.py attention_qkv_prob_dropout)."""

    def __init__(self, cfg: BertConfig):
        super().__init__()
        import os
        self.heads = cfg.heads
        self.head_dim = cfg.hidden // cfg.heads
        self.qkv = make_linear(cfg.hidden, 3 * cfg.hidden)
        exact = os.environ.get("RAVNEST_EXACT_ATTN_DROPOUT", "0") == "1"
        self.core = AttentionCoreQKV(
            causal=False, prob_dropout=cfg.dropout if exact else 0.0)
        self.out = make_linear(cfg.hidden, cfg.hidden)
        self.drop = Dropout(cfg.dropout)

    def forward(self, x, mask):
        # packed (B,S,3,H,D) qkv: the attention kernel reads the
        # projection's natural layout and returns token-major O — no
        # permute/contiguous copies (unflatten also keeps the fx graph
        # free of .size() scalar nodes)
        qkv = self.qkv(x).unflatten(-1, (3, self.heads, self.head_dim))
        o = self.core(qkv, mask)
        return self.drop(self.out(o))


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.attn = BertSelfAttention(cfg)
        self.ln1 = AddLayerNorm(cfg.hidden, cfg.layer_norm_eps)
        self.mlp_in = LinearGelu(cfg.hidden, cfg.intermediate)
        self.mlp_out = make_linear(cfg.intermediate, cfg.hidden)
        self.drop = Dropout(cfg.dropout)
        self.ln2 = AddLayerNorm(cfg.hidden, cfg.layer_norm_eps)

    def forward(self, x, mask):
        x = self.ln1(x, self.attn(x, mask))
        x = self.ln2(x, self.drop(self.mlp_out(self.mlp_in(x))))
        return x


class BertMLMHead(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.dense = make_linear(cfg.hidden, cfg.hidden)
        self.act = GELU()
        self.ln = FusedLayerNorm(cfg.hidden, cfg.layer_norm_eps)
        self.decoder = make_linear(cfg.hidden, cfg.vocab_size)

    def forward(self, x):
        return self.decoder(self.ln(self.act(self.dense(x))))


class BertForMLM(nn.Module):
    """input_ids (B,S) int64, attention_mask (B,S) {0,1} -> logits (B,S,V)."""

    def __init__(self, cfg: BertConfig | None = None):
        super().__init__()
        self.cfg = cfg or BertConfig.base()
        self.embeddings = BertEmbeddings(self.cfg)
        self.layers = nn.ModuleList(
            [BertLayer(self.cfg) for _ in range(self.cfg.layers)])
        self.head = BertMLMHead(self.cfg)
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

    def forward(self, input_ids, attention_mask):
        # additive mask: (B,S) {0,1} -> (B,1,1,S) {0,-inf-ish}
        m = (1.0 - attention_mask.to(torch.float32)) * -10000.0
        m = m.unsqueeze(1).unsqueeze(2)
        x = self.embeddings(input_ids)
        for layer in self.layers:
            x = layer(x, m)
        return self.head(x)
