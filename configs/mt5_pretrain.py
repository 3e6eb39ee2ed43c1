"""MT5-style pretraining: T5 with the gated-gelu MLP (reference:
projects/MT5 -- fused_fast_gelu_mul gated MLP, shared enc/dec embedding)."""

from .t5_pretrain import dataloader, model, optim, train  # noqa: F401

# mutate the model's own cfg node (each config import gets its own module
# instance, so the common-file `cfg` alias would be a different object)
model.cfg.mlp_type = "gated"
model.cfg.activation = "gelu"
# MT5/T5.1.1 uses bucketed relative-position biases instead of absolute
# positions (reference projects/MT5/layers/attention_layer.py:118-123)
model.cfg.relative_attention = True

train.update(output_dir="./output/mt5_pretrain")
