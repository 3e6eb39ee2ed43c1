from .build import build_optimizer, get_default_optimizer_params, reduce_param_groups
from .fused_adamw import FusedAdamW

__all__ = [
    "FusedAdamW",
    "build_optimizer",
    "get_default_optimizer_params",
    "reduce_param_groups",
]
