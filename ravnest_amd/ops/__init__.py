from ._ext import get_ext, has_ext
from .layernorm import FusedLayerNorm, layer_norm
from .activation import GELU, LinearGelu, gelu, bias_gelu
from .dropout import Dropout, dropout
from .attention import AttentionCore, attention
from .losses import CrossEntropyLoss, cross_entropy
from .optim import FusedAdam, FusedSGD, FusedLAMB

__all__ = [
    "get_ext", "has_ext",
    "FusedLayerNorm", "layer_norm",
    "GELU", "LinearGelu", "gelu", "bias_gelu",
    "Dropout", "dropout",
    "AttentionCore", "attention",
    "CrossEntropyLoss", "cross_entropy",
    "FusedAdam", "FusedSGD", "FusedLAMB",
]
