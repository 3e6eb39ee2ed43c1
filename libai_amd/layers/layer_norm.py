"""LayerNorm / RMSLayerNorm modules over the HIP norm kernels.

Reference: libai/layers/layer_norm.py:78-131 (flow._C.layer_norm_affine /
flow._C.rms_norm).  Affine params are replicated across TP.
"""

import torch
from torch import nn

from ..ops.norm import layer_norm, rms_norm

__all__ = ["LayerNorm", "RMSLayerNorm", "RMSNorm"]


class LayerNorm(nn.Module):
    def __init__(self, normalized_shape, eps=1e-5, *, layer_idx=0, dtype=None,
                 elementwise_affine=True, bias=True):
        super().__init__()
        if isinstance(normalized_shape, int):
            normalized_shape = (normalized_shape,)
        assert len(normalized_shape) == 1, "only last-dim LayerNorm is supported"
        self.normalized_shape = tuple(normalized_shape)
        self.eps = eps
        self.layer_idx = layer_idx
        dtype = dtype or torch.get_default_dtype()
        if elementwise_affine:
            self.weight = nn.Parameter(torch.ones(normalized_shape, dtype=dtype))
            if bias:
                self.bias = nn.Parameter(torch.zeros(normalized_shape, dtype=dtype))
            else:
                self.register_parameter("bias", None)
        else:
            self.register_parameter("weight", None)
            self.register_parameter("bias", None)

    def forward(self, x):
        if self.weight is None:
            return torch.nn.functional.layer_norm(x, self.normalized_shape, eps=self.eps)
        return layer_norm(x, self.weight, self.bias, self.eps)


class RMSLayerNorm(nn.Module):
    def __init__(self, normalized_shape, eps=1e-6, *, layer_idx=0, dtype=None):
        super().__init__()
        if isinstance(normalized_shape, int):
            normalized_shape = (normalized_shape,)
        self.normalized_shape = tuple(normalized_shape)
        self.eps = eps
        self.layer_idx = layer_idx
        dtype = dtype or torch.get_default_dtype()
        self.weight = nn.Parameter(torch.ones(normalized_shape, dtype=dtype))

    def forward(self, x):
        return rms_norm(x, self.weight, self.eps)


RMSNorm = RMSLayerNorm
