"""Sequence parallelism: ring attention over sharded sequence chunks.

The reference has NO sequence/context parallelism (SURVEY.md section 2.2
row SP: absent; section 5 asks the MI355X build to leave room for it in
the comm design). This package adds it as an orthogonal axis on top of
the existing kernels: the flash attention kernel already returns the
per-row log-sum-exp, so ring attention composes at the tensor level —
no new device code needed (csrc/attention.hip fwd, attention_bwd.hip
bwd with a caller-provided GLOBAL lse).
"""
from .ring_attention import (RingAttention, ring_attention,
                             merge_partials)
from .sp import (RingCoreQKV, allreduce_gradients,
                 sequence_parallelize)

__all__ = ["RingAttention", "ring_attention", "merge_partials",
           "RingCoreQKV", "sequence_parallelize", "allreduce_gradients"]
