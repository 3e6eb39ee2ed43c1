from ._ext import draw_seed, ext, has_ext, use_hip
from .attention import flash_attention, flash_attention_available
from .cross_entropy import vocab_parallel_cross_entropy
from .fused_bias import bias_dropout_add, bias_gelu
from .norm import layer_norm, rms_norm
from .rope import RotaryEmbedding, apply_rotary_pos_emb
from .softmax import fused_scale_mask_softmax
from .swiglu import swiglu

__all__ = [
    "ext",
    "has_ext",
    "use_hip",
    "draw_seed",
    "layer_norm",
    "rms_norm",
    "bias_gelu",
    "bias_dropout_add",
    "fused_scale_mask_softmax",
    "vocab_parallel_cross_entropy",
    "flash_attention",
    "flash_attention_available",
    "apply_rotary_pos_emb",
    "RotaryEmbedding",
    "swiglu",
]
