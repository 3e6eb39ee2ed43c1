"""Activation registry (reference: libai/layers/activation.py:71-87)."""

import torch
import torch.nn.functional as F
from torch import nn

__all__ = ["build_activation"]


class QuickGELU(nn.Module):
    def forward(self, x):
        return x * torch.sigmoid(1.702 * x)


class SquaredReLU(nn.Module):
    def forward(self, x):
        r = F.relu(x)
        return r * r


_ACTIVATIONS = {
    "gelu": nn.GELU,
    "gelu_tanh": lambda: nn.GELU(approximate="tanh"),
    "tanh": nn.Tanh,
    "relu": nn.ReLU,
    "quick_gelu": QuickGELU,
    "squared_relu": SquaredReLU,
    "sigmoid": nn.Sigmoid,
    "silu": nn.SiLU,
}


def build_activation(name):
    if name is None:
        name = "gelu"
    try:
        return _ACTIVATIONS[name]()
    except KeyError:
        raise KeyError(
            f"unknown activation {name!r}; available: {sorted(_ACTIVATIONS)}"
        ) from None
