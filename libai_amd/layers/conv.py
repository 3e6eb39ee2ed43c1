"""GPT-2-style Conv1D: a linear with transposed weight storage [in, out].

Reference: libai/layers/conv.py:29+ (same col/row TP logic as Linear1D).
"""

import torch
from torch import nn

from ..parallel.comm import (
    copy_to_tensor_parallel_region,
    reduce_from_tensor_parallel_region,
)
from ..utils import distributed as du
from .linear import init_tp_shard_

__all__ = ["Conv1D"]


class Conv1D(nn.Module):
    def __init__(self, in_features, out_features, bias=True, parallel="data",
                 init_method=nn.init.xavier_normal_, skip_bias_add=False,
                 *, layer_idx=0, dtype=None):
        super().__init__()
        assert parallel in ("data", "col", "row")
        self.in_features = in_features
        self.out_features = out_features
        self.parallel = parallel
        self.skip_bias_add = skip_bias_add
        self.layer_idx = layer_idx
        dutil = du.get_dist_util()
        tp = dutil.tensor_parallel_size
        dtype = dtype or torch.get_default_dtype()

        if parallel == "col":
            assert out_features % tp == 0
            self.weight = nn.Parameter(torch.empty(in_features, out_features // tp, dtype=dtype))
            self.weight.tensor_parallel = True
            self.weight.tp_shard_dim = 1
            init_tp_shard_(self.weight, (in_features, out_features), init_method, 1)
            self.bias = (
                nn.Parameter(torch.zeros(out_features // tp, dtype=dtype)) if bias else None
            )
            if self.bias is not None:
                self.bias.tensor_parallel = True
                self.bias.tp_shard_dim = 0
        elif parallel == "row":
            assert in_features % tp == 0
            self.weight = nn.Parameter(torch.empty(in_features // tp, out_features, dtype=dtype))
            self.weight.tensor_parallel = True
            self.weight.tp_shard_dim = 0
            init_tp_shard_(self.weight, (in_features, out_features), init_method, 0)
            self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype)) if bias else None
        else:
            self.weight = nn.Parameter(torch.empty(in_features, out_features, dtype=dtype))
            init_method(self.weight.data)
            self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype)) if bias else None
        if self.bias is None:
            self.register_parameter("bias", None)

    def forward(self, x):
        if self.parallel == "col":
            x = copy_to_tensor_parallel_region(x)
            out = torch.matmul(x, self.weight)
        elif self.parallel == "row":
            out = torch.matmul(x, self.weight)
            out = reduce_from_tensor_parallel_region(out)
        else:
            out = torch.matmul(x, self.weight)
        if self.skip_bias_add:
            return out, self.bias
        if self.bias is not None:
            out = out + self.bias
        return out
