from .activation import build_activation
from .attention import AttnMaskType, MultiheadAttention
from .conv import Conv1D
from .cross_entropy import ParallelCrossEntropyLoss
from .droppath import DropPath, drop_path
from .embedding import Embedding, PatchEmbedding, SinePositionalEmbedding, VocabEmbedding
from .layer_norm import LayerNorm, RMSLayerNorm, RMSNorm
from .linear import Linear, Linear1D
from .lm_logits import LMLogits
from .moe import MoELayer
from .position_bias import T5RelativePositionBias, build_alibi_bias
from .mlp import MLP
from .transformer_layer import TransformerLayer

__all__ = [
    "build_activation",
    "AttnMaskType",
    "MultiheadAttention",
    "Conv1D",
    "ParallelCrossEntropyLoss",
    "DropPath",
    "drop_path",
    "Embedding",
    "PatchEmbedding",
    "SinePositionalEmbedding",
    "VocabEmbedding",
    "LayerNorm",
    "RMSLayerNorm",
    "RMSNorm",
    "Linear",
    "Linear1D",
    "LMLogits",
    "MLP",
    "MoELayer",
    "T5RelativePositionBias",
    "build_alibi_bias",
    "TransformerLayer",
]
