"""Transformer MLP: col-parallel h->4h, fused bias-gelu, row-parallel 4h->h.

Reference: libai/layers/mlp.py:65-108 (flow._C.fused_bias_add_gelu +
fused_bias_add_dropout call sites).  The output bias/dropout/residual is
fused by the caller via ``residual=``, like MultiheadAttention.
"""

from torch import nn

from ..ops.fused_bias import bias_dropout_add, bias_gelu
from .activation import build_activation
from .linear import Linear1D

__all__ = ["MLP"]


class MLP(nn.Module):
    def __init__(
        self,
        hidden_size,
        ffn_hidden_size,
        output_dropout_prob=0.0,
        init_method=nn.init.xavier_normal_,
        output_layer_init_method=None,
        bias_gelu_fusion=True,
        bias_dropout_fusion=True,
        activation="gelu",
        *,
        layer_idx=0,
    ):
        super().__init__()
        self.output_dropout_prob = output_dropout_prob
        self.layer_idx = layer_idx
        self.bias_gelu_fusion = bias_gelu_fusion and activation == "gelu"
        output_layer_init_method = output_layer_init_method or init_method

        self.dense_h_to_4h = Linear1D(
            hidden_size, ffn_hidden_size, parallel="col", init_method=init_method,
            skip_bias_add=self.bias_gelu_fusion, layer_idx=layer_idx,
        )
        self.dense_4h_to_h = Linear1D(
            ffn_hidden_size, hidden_size, parallel="row",
            init_method=output_layer_init_method, skip_bias_add=True,
            layer_idx=layer_idx,
        )
        self.activation_func = None if self.bias_gelu_fusion else build_activation(activation)

    def forward(self, hidden_states, residual=None):
        if self.bias_gelu_fusion:
            inter, bias = self.dense_h_to_4h(hidden_states)
            inter = bias_gelu(inter, bias)
        else:
            inter = self.activation_func(self.dense_h_to_4h(hidden_states))
        out, bias = self.dense_4h_to_h(inter)
        return bias_dropout_add(
            out, bias=bias, residual=residual, p=self.output_dropout_prob,
            training=self.training,
        )
