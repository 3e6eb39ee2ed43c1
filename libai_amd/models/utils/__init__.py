from .weight_init import init_method_normal, scaled_init_method_normal

__all__ = ["init_method_normal", "scaled_init_method_normal"]
