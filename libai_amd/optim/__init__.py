from .fused_adamw import FusedAdamW

__all__ = ["FusedAdamW"]
