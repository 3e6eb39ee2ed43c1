"""Pre-LN transformer block (reference: libai/layers/transformer_layer.py:130-232).

Structure: LN -> attention (+fused residual) -> LN -> [cross-attn] -> MLP
(+fused residual).  ``layer_idx`` drives the pipeline-stage assignment; the
pipeline engine (libai_amd/parallel/pipeline.py) moves activations between
stages — layers never do (the reference's to_global hop at
transformer_layer.py:158 becomes an explicit P2P in the engine).
"""

import torch
from torch import nn

from .attention import AttnMaskType, MultiheadAttention
from .layer_norm import LayerNorm
from .mlp import MLP

__all__ = ["TransformerLayer"]


class TransformerLayer(nn.Module):
    def __init__(
        self,
        hidden_size,
        ffn_hidden_size,
        num_attention_heads,
        is_decoder=False,
        attention_dropout_prob=0.0,
        output_dropout_prob=0.0,
        layernorm_epsilon=1e-5,
        init_method=nn.init.xavier_normal_,
        output_layer_init_method=None,
        bias_gelu_fusion=True,
        bias_dropout_fusion=True,
        scale_mask_softmax_fusion=True,
        apply_query_key_layer_scaling=False,
        apply_residual_post_layernorm=False,
        attn_mask_type=AttnMaskType.padding,
        mlp_type="dense",
        activation="gelu",
        sequence_parallel=False,
        moe_num_experts=8,
        moe_top_k=2,
        *,
        layer_idx=0,
    ):
        super().__init__()
        self.layer_idx = layer_idx
        self.is_decoder = is_decoder
        self.apply_residual_post_layernorm = apply_residual_post_layernorm
        self.sequence_parallel = sequence_parallel
        assert not (sequence_parallel and is_decoder), \
            "sequence parallelism: encoder/causal self-attention blocks only"
        assert not (sequence_parallel and mlp_type == "gated"), \
            "sequence parallelism + gated MLP not wired yet"
        output_layer_init_method = output_layer_init_method or init_method
        self._sp = sequence_parallel

        self.input_layernorm = LayerNorm(hidden_size, eps=layernorm_epsilon,
                                         layer_idx=layer_idx)
        self.self_attention = MultiheadAttention(
            hidden_size, num_attention_heads,
            attention_dropout_prob=attention_dropout_prob,
            output_dropout_prob=output_dropout_prob,
            init_method=init_method,
            output_layer_init_method=output_layer_init_method,
            bias_dropout_fusion=bias_dropout_fusion,
            scale_mask_softmax_fusion=scale_mask_softmax_fusion,
            apply_query_key_layer_scaling=apply_query_key_layer_scaling,
            attn_mask_type=attn_mask_type,
            sequence_parallel=sequence_parallel,
            layer_idx=layer_idx,
        )
        self.post_attention_layernorm = LayerNorm(hidden_size, eps=layernorm_epsilon,
                                                  layer_idx=layer_idx)
        if is_decoder:
            self.cross_attention = MultiheadAttention(
                hidden_size, num_attention_heads, is_cross_attention=True,
                attention_dropout_prob=attention_dropout_prob,
                output_dropout_prob=output_dropout_prob,
                init_method=init_method,
                output_layer_init_method=output_layer_init_method,
                attn_mask_type=AttnMaskType.padding,
                layer_idx=layer_idx,
            )
            self.post_cross_attention_layernorm = LayerNorm(
                hidden_size, eps=layernorm_epsilon, layer_idx=layer_idx
            )
        if mlp_type == "moe":
            from .moe import MoELayer

            assert not sequence_parallel, "MoE + SP not wired yet"
            self.mlp = MoELayer(
                hidden_size, ffn_hidden_size, num_experts=moe_num_experts,
                top_k=moe_top_k, activation=activation,
                init_method=init_method,
                output_layer_init_method=output_layer_init_method,
                layer_idx=layer_idx,
            )
        elif mlp_type == "gated":
            from .mlp import GatedMLP

            self.mlp = GatedMLP(
                hidden_size, ffn_hidden_size,
                output_dropout_prob=output_dropout_prob,
                activation=activation,
                init_method=init_method,
                output_layer_init_method=output_layer_init_method,
                layer_idx=layer_idx,
            )
        else:
            self.mlp = MLP(
                hidden_size, ffn_hidden_size,
                output_dropout_prob=output_dropout_prob,
                init_method=init_method,
                output_layer_init_method=output_layer_init_method,
                bias_gelu_fusion=bias_gelu_fusion,
                bias_dropout_fusion=bias_dropout_fusion,
                activation=activation,
                sequence_parallel=sequence_parallel,
                layer_idx=layer_idx,
            )
        if sequence_parallel:
            # LayerNorms run on seq shards: their affine grads are partial
            # sums over local tokens and need a TP all-reduce at grad sync
            for ln in (self.input_layernorm, self.post_attention_layernorm):
                for p in ln.parameters():
                    p.sequence_parallel_grad = True

    def forward(
        self,
        hidden_states,
        attention_mask=None,
        encoder_states=None,
        encoder_attention_mask=None,
        past_key_value=None,
        use_cache=False,
        position_bias=None,
        static_cache=None,
        position=None,
    ):
        if static_cache is not None:
            # hipGraph-capturable decode step: fixed-shape path, no cache
            # tuples returned (KV written in-place into the static buffers);
            # bias + residual + post-attention norm run as ONE kernel
            from ..ops._ext import ext
            from ..ops.fused_bias import bias_dropout_add

            ln1 = self.input_layernorm(hidden_states)
            residual = ln1 if self.apply_residual_post_layernorm else hidden_states
            out, bias = self.self_attention(ln1, residual=residual,
                                            static_cache=static_cache,
                                            position=position)
            ln = self.post_attention_layernorm
            if out.shape[-1] <= 2048:
                h, ln2 = ext().res_norm_fwd(out, bias, residual, ln.weight,
                                            ln.bias, ln.eps, False)
            else:  # wide rows exceed the wave kernel's register budget
                h = bias_dropout_add(out, bias=bias, residual=residual,
                                     p=0.0, training=False)
                ln2 = ln(h)
            residual = ln2 if self.apply_residual_post_layernorm else h
            return self.mlp(ln2, residual=residual)
        if past_key_value is not None:
            if self.is_decoder:
                self_past, cross_past = past_key_value
            else:
                self_past, cross_past = past_key_value, None
        else:
            self_past, cross_past = None, None

        ln1 = self.input_layernorm(hidden_states)
        residual = ln1 if self.apply_residual_post_layernorm else hidden_states
        attn_out = self.self_attention(
            ln1, attention_mask=attention_mask, past_key_value=self_past,
            use_cache=use_cache, residual=residual, position_bias=position_bias,
        )
        if use_cache:
            attn_out, self_present = attn_out
        hidden_states = attn_out

        presents = None
        if self.is_decoder and encoder_states is not None:
            ln_c = self.post_attention_layernorm(hidden_states)
            residual = ln_c if self.apply_residual_post_layernorm else hidden_states
            cross_out = self.cross_attention(
                ln_c, encoder_states=encoder_states,
                attention_mask=encoder_attention_mask, past_key_value=cross_past,
                use_cache=use_cache, residual=residual,
            )
            if use_cache:
                cross_out, cross_present = cross_out
                presents = (self_present, cross_present)
            hidden_states = cross_out
            ln2 = self.post_cross_attention_layernorm(hidden_states)
        else:
            if use_cache:
                presents = self_present
            ln2 = self.post_attention_layernorm(hidden_states)

        residual = ln2 if self.apply_residual_post_layernorm else hidden_states
        out = self.mlp(ln2, residual=residual)
        if use_cache:
            return out, presents
        return out
