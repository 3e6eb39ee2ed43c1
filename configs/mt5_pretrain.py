"""MT5-style pretraining: T5 with the gated-gelu MLP (reference:
projects/MT5 -- fused_fast_gelu_mul gated MLP, shared enc/dec embedding)."""

from .common.models.t5 import cfg as t5_cfg
from .t5_pretrain import dataloader, model, optim, train  # noqa: F401

t5_cfg.mlp_type = "gated"
t5_cfg.activation = "gelu"

train.update(output_dir="./output/mt5_pretrain")
