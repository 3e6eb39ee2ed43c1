"""PaLM-style decoder: PARALLEL attention + MLP from one LayerNorm
(y = x + attn(ln(x)) + mlp(ln(x))), multi-query attention, SwiGLU, RoPE.

Reference capability: projects/PaLM (PaLM model definition on the library).
The parallel formulation halves the LayerNorm count and lets the attention
and MLP projections share the normalized input; multi-query attention is
the kv_heads=1 extreme of the GQA machinery.
"""

import torch
from torch import nn
from torch.utils.checkpoint import checkpoint as act_checkpoint

from ..config import configurable
from ..layers import LMLogits, ParallelCrossEntropyLoss, RMSLayerNorm, VocabEmbedding
from .llama import LlamaAttention, LlamaMLP
from .utils.weight_init import init_method_normal, scaled_init_method_normal

__all__ = ["PaLMModel", "PaLMForCausalLM"]


class PaLMBlock(nn.Module):
    def __init__(self, hidden_size, intermediate_size, num_heads,
                 max_position_embeddings, rms_norm_eps, init_method,
                 output_init_method, num_key_value_heads=1, *, layer_idx=0):
        super().__init__()
        self.layer_idx = layer_idx
        self.norm = RMSLayerNorm(hidden_size, eps=rms_norm_eps,
                                 layer_idx=layer_idx)
        self.attn = LlamaAttention(hidden_size, num_heads,
                                   max_position_embeddings, init_method,
                                   output_init_method,
                                   num_key_value_heads=num_key_value_heads,
                                   layer_idx=layer_idx)
        self.mlp = LlamaMLP(hidden_size, intermediate_size, init_method,
                            output_init_method, layer_idx=layer_idx)

    def forward(self, x, past_key_value=None, use_cache=False):
        ln = self.norm(x)
        a = self.attn(ln, past_key_value=past_key_value, use_cache=use_cache)
        if use_cache:
            a, present = a
        m = self.mlp(ln)
        out = x + a + m  # parallel residual combine
        if use_cache:
            return out, present
        return out


class PaLMModel(nn.Module):
    @configurable
    def __init__(
        self,
        hidden_layers,
        vocab_size,
        hidden_size,
        intermediate_size,
        num_attention_heads,
        num_key_value_heads=1,  # PaLM uses multi-query attention
        max_position_embeddings=2048,
        rms_norm_eps=1e-5,
        initializer_range=0.02,
        amp_enabled=False,
    ):
        super().__init__()
        init_method = init_method_normal(initializer_range)
        output_init = scaled_init_method_normal(initializer_range, hidden_layers)
        self.embed_tokens = VocabEmbedding(vocab_size, hidden_size,
                                           init_method=init_method, layer_idx=0)
        self.layers = nn.ModuleList([
            PaLMBlock(hidden_size, intermediate_size, num_attention_heads,
                      max_position_embeddings, rms_norm_eps, init_method,
                      output_init, num_key_value_heads, layer_idx=i)
            for i in range(hidden_layers)
        ])
        self.norm = RMSLayerNorm(hidden_size, eps=rms_norm_eps, layer_idx=-1)
        self.hidden_size = hidden_size
        self.checkpoint_activations = False

    @classmethod
    def from_config(cls, cfg):
        return {k: cfg.get(k) for k in (
            "hidden_layers", "vocab_size", "hidden_size", "intermediate_size",
            "num_attention_heads", "num_key_value_heads",
            "max_position_embeddings", "rms_norm_eps", "initializer_range",
        ) if cfg.get(k) is not None}

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        h = self.embed_tokens(input_ids)
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            if self.checkpoint_activations and self.training and not use_cache:
                h = act_checkpoint(layer, h, use_reentrant=False)
            else:
                h = layer(h, past_key_value=past, use_cache=use_cache)
            if use_cache:
                h, p = h
                presents.append(p)
        h = self.norm(h)
        if use_cache:
            return h, presents
        return h

    def set_activation_checkpoint(self, enabled=True):
        self.checkpoint_activations = enabled


class PaLMForCausalLM(nn.Module):
    @configurable
    def __init__(self, cfg=None, **kwargs):
        super().__init__()
        self.model = PaLMModel(cfg) if cfg is not None else PaLMModel(**kwargs)
        vocab = cfg.vocab_size if cfg is not None else kwargs["vocab_size"]
        self.lm_logits = LMLogits(vocab, bias=False, layer_idx=-1)
        self.loss_func = ParallelCrossEntropyLoss()
        self.hidden_size = self.model.hidden_size

    @classmethod
    def from_config(cls, cfg):
        return {"cfg": cfg}

    def forward(self, input_ids, labels=None, past_key_values=None,
                use_cache=False):
        h = self.model(input_ids, past_key_values=past_key_values,
                       use_cache=use_cache)
        if use_cache:
            h, presents = h
        # PaLM ties the output projection to the input embedding
        logits = self.lm_logits(h, self.model.embed_tokens.weight)
        if labels is not None:
            return {"lm_loss": self.loss_func(logits, labels).mean()}
        if use_cache:
            return {"prediction_scores": logits, "past_key_values": presents}
        return {"prediction_scores": logits}

    def set_activation_checkpoint(self, enabled=True):
        self.model.set_activation_checkpoint(enabled)
