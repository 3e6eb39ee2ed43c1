"""Megatron-style init helpers (reference: libai/models/utils/weight_init.py)."""

import math

from torch import nn

__all__ = ["init_method_normal", "scaled_init_method_normal"]


def init_method_normal(sigma):
    def init_(tensor):
        return nn.init.normal_(tensor, mean=0.0, std=sigma)

    return init_


def scaled_init_method_normal(sigma, num_layers):
    """std scaled by 1/sqrt(2*num_layers) for output-projection weights."""
    std = sigma / math.sqrt(2.0 * num_layers)

    def init_(tensor):
        return nn.init.normal_(tensor, mean=0.0, std=std)

    return init_
