from ._ext import get_ext, has_ext
from .layernorm import (FusedLayerNorm, AddLayerNorm, layer_norm, add_layer_norm)
from .activation import (GELU, LinearGelu, Softmax, gelu, bias_gelu,
                         softmax)
from .dropout import Dropout, dropout
from .attention import (AttentionCore, AttentionCoreQKV, attention, attention_qkv)
from .losses import CrossEntropyLoss, cross_entropy
from .optim import FusedAdam, FusedSGD, FusedLAMB
from .batchnorm import FusedBatchNorm2d
from .linear import Linear
from .conv import Conv1x1, conv2d_mfma
from .embedding import embedding_ln, embedding_add
from .pool import MaxPool2d, AvgPool2d, AdaptiveAvgPool2d
from .mx import MXLinear, mx_linear

__all__ = [
    "get_ext", "has_ext",
    "FusedLayerNorm", "AddLayerNorm", "layer_norm", "add_layer_norm",
    "GELU", "LinearGelu", "Softmax", "gelu", "bias_gelu", "softmax",
    "Dropout", "dropout",
    "AttentionCore", "AttentionCoreQKV", "attention", "attention_qkv",
    "CrossEntropyLoss", "cross_entropy",
    "FusedAdam", "FusedSGD", "FusedLAMB",
    "FusedBatchNorm2d", "Linear", "Conv1x1", "conv2d_mfma",
    "embedding_ln", "embedding_add",
    "MaxPool2d", "AvgPool2d", "AdaptiveAvgPool2d",
    "MXLinear", "mx_linear",
]
