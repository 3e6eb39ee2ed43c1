"""Transformer MLP: col-parallel h->4h, fused bias-gelu, row-parallel 4h->h.

Reference: libai/layers/mlp.py:65-108 (flow._C.fused_bias_add_gelu +
fused_bias_add_dropout call sites).  The output bias/dropout/residual is
fused by the caller via ``residual=``, like MultiheadAttention.
"""

import torch
from torch import nn

from ..ops.fused_bias import bias_dropout_add, bias_gelu
from .activation import build_activation
from .linear import Linear1D

__all__ = ["MLP", "GatedMLP"]


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
        sequence_parallel=False,
        *,
        layer_idx=0,
    ):
        super().__init__()
        self.output_dropout_prob = output_dropout_prob
        self.layer_idx = layer_idx
        self.bias_gelu_fusion = bias_gelu_fusion and activation == "gelu"
        self.sequence_parallel = sequence_parallel
        output_layer_init_method = output_layer_init_method or init_method

        self.dense_h_to_4h = Linear1D(
            hidden_size, ffn_hidden_size, parallel="col", init_method=init_method,
            skip_bias_add=self.bias_gelu_fusion,
            sequence_parallel=sequence_parallel, layer_idx=layer_idx,
        )
        self.dense_4h_to_h = Linear1D(
            ffn_hidden_size, hidden_size, parallel="row",
            init_method=output_layer_init_method, skip_bias_add=True,
            sequence_parallel=sequence_parallel, layer_idx=layer_idx,
        )
        self.activation_func = None if self.bias_gelu_fusion else build_activation(activation)

    def forward(self, hidden_states, residual=None):
        if self.bias_gelu_fusion:
            from ..ops.fused_mlp import fused_mlp, fused_mlp_available

            if fused_mlp_available(hidden_states):
                # epilogue-fused: gelu(xW1+b1) in the GEMM, dgelu+db1 in the
                # backward GEMM — no separate elementwise kernels
                from ..parallel.comm import (
                    copy_to_tensor_parallel_region,
                    gather_from_sequence_parallel_region,
                    reduce_from_tensor_parallel_region,
                    reduce_scatter_to_sequence_parallel_region,
                )

                if self.sequence_parallel:
                    x = gather_from_sequence_parallel_region(hidden_states)
                else:
                    x = copy_to_tensor_parallel_region(hidden_states)
                out = fused_mlp(x, self.dense_h_to_4h.weight,
                                self.dense_h_to_4h.bias,
                                self.dense_4h_to_h.weight)
                if self.sequence_parallel:
                    out = reduce_scatter_to_sequence_parallel_region(out)
                else:
                    out = reduce_from_tensor_parallel_region(out)
                return bias_dropout_add(
                    out, bias=self.dense_4h_to_h.bias, residual=residual,
                    p=self.output_dropout_prob, training=self.training,
                )
            inter, bias = self.dense_h_to_4h(hidden_states)
            inter = bias_gelu(inter, bias)
        else:
            inter = self.activation_func(self.dense_h_to_4h(hidden_states))
        out, bias = self.dense_4h_to_h(inter)
        return bias_dropout_add(
            out, bias=bias, residual=residual, p=self.output_dropout_prob,
            training=self.training,
        )


class GatedMLP(nn.Module):
    """MT5/GLM-style gated MLP: out = W2 (act(x W1) * (x W3)), bias-free.

    Reference: projects/MT5 fused_fast_gelu_mul (mlp_layer.py:123) and the
    Llama SwiGLU variant.  The fused [gate|up] column
    projection keeps one GEMM on the 2*ffn width; activation "silu" uses
    the fused SwiGLU HIP kernel, "gelu" composes gelu(gate)*up.
    """

    def __init__(self, hidden_size, ffn_hidden_size, output_dropout_prob=0.0,
                 activation="gelu", init_method=nn.init.xavier_normal_,
                 output_layer_init_method=None, *, layer_idx=0):
        super().__init__()
        self.output_dropout_prob = output_dropout_prob
        self.activation = activation
        output_layer_init_method = output_layer_init_method or init_method
        self.gate_up_proj = Linear1D(
            hidden_size, 2 * ffn_hidden_size, bias=False, parallel="col",
            init_method=init_method, fused_chunks=2, layer_idx=layer_idx,
        )
        self.down_proj = Linear1D(
            ffn_hidden_size, hidden_size, bias=False, parallel="row",
            init_method=output_layer_init_method, skip_bias_add=True,
            layer_idx=layer_idx,
        )

    def forward(self, hidden_states, residual=None):
        gu = self.gate_up_proj(hidden_states)
        if self.activation == "silu":
            from ..ops.swiglu import swiglu

            inter = swiglu(gu)
        else:
            gate, up = gu.chunk(2, dim=-1)
            inter = torch.nn.functional.gelu(gate, approximate="tanh") * up
        out, _ = self.down_proj(inter)
        return bias_dropout_add(
            out, bias=None, residual=residual, p=self.output_dropout_prob,
            training=self.training,
        )
