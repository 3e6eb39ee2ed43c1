"""BLOOM causal LM: ALiBi attention, embedding LayerNorm, tied logits.

Reference capability: projects/BLOOM/modeling/ (alibi attention, 2D tp+pp
inference) built on the library layers.  Here BLOOM is a first-class model:
the ALiBi slopes feed TransformerLayer's ``position_bias`` and everything
else reuses the TP/PP-aware layer stack.
"""

import torch
from torch import nn
from torch.utils.checkpoint import checkpoint as act_checkpoint

from ..config import configurable
from ..layers import (
    AttnMaskType,
    LayerNorm,
    LMLogits,
    ParallelCrossEntropyLoss,
    TransformerLayer,
    VocabEmbedding,
)
from ..layers.position_bias import build_alibi_bias
from .utils.weight_init import init_method_normal, scaled_init_method_normal

__all__ = ["BloomModel", "BloomForCausalLM"]


class BloomModel(nn.Module):
    @configurable
    def __init__(
        self,
        vocab_size,
        hidden_size,
        hidden_layers,
        num_attention_heads,
        ffn_hidden_size=None,
        hidden_dropout_prob=0.0,
        attention_dropout_prob=0.0,
        initializer_range=0.02,
        layernorm_eps=1e-5,
        apply_residual_post_layernorm=False,
    ):
        super().__init__()
        init_method = init_method_normal(initializer_range)
        scaled_init = scaled_init_method_normal(initializer_range, hidden_layers)
        self.hidden_size = hidden_size
        self.num_heads = num_attention_heads
        ffn_hidden_size = ffn_hidden_size or 4 * hidden_size

        self.word_embeddings = VocabEmbedding(vocab_size, hidden_size,
                                              init_method=init_method, layer_idx=0)
        # BLOOM normalizes the embedding output (no positional embeddings —
        # ALiBi carries all position information)
        self.word_embeddings_layernorm = LayerNorm(hidden_size, eps=layernorm_eps,
                                                   layer_idx=0)
        self.layers = nn.ModuleList([
            TransformerLayer(
                hidden_size, ffn_hidden_size, num_attention_heads,
                attention_dropout_prob=attention_dropout_prob,
                output_dropout_prob=hidden_dropout_prob,
                layernorm_epsilon=layernorm_eps,
                init_method=init_method,
                output_layer_init_method=scaled_init,
                attn_mask_type=AttnMaskType.causal,
                apply_residual_post_layernorm=apply_residual_post_layernorm,
                layer_idx=i,
            )
            for i in range(hidden_layers)
        ])
        self.ln_f = LayerNorm(hidden_size, eps=layernorm_eps, layer_idx=-1)
        self.lm_head = LMLogits(vocab_size, bias=False, layer_idx=-1)
        self.checkpoint_activations = False
        self._alibi_cache = {}

    @classmethod
    def from_config(cls, cfg):
        return {
            "vocab_size": cfg.vocab_size,
            "hidden_size": cfg.hidden_size,
            "hidden_layers": cfg.hidden_layers,
            "num_attention_heads": cfg.num_attention_heads,
            "ffn_hidden_size": cfg.get("ffn_hidden_size", None),
            "hidden_dropout_prob": cfg.get("hidden_dropout_prob", 0.0),
            "attention_dropout_prob": cfg.get("attention_dropout_prob", 0.0),
            "initializer_range": cfg.get("initializer_range", 0.02),
            "layernorm_eps": cfg.get("layernorm_eps", 1e-5),
            "apply_residual_post_layernorm": cfg.get(
                "apply_residual_post_layernorm", False),
        }

    def alibi(self, seq_len, device, dtype):
        key = (seq_len, device, dtype)
        if key not in self._alibi_cache:
            self._alibi_cache[key] = build_alibi_bias(
                self.num_heads, seq_len, device=device, dtype=dtype)
        return self._alibi_cache[key]

    def _run(self, layer, *args, **kw):
        if self.checkpoint_activations and self.training:
            return act_checkpoint(layer, *args, use_reentrant=False, **kw)
        return layer(*args, **kw)

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        past_len = (
            past_key_values[0][0].shape[2] if past_key_values is not None else 0
        )
        h = self.word_embeddings_layernorm(self.word_embeddings(input_ids))
        # keys cover [0, past+s); the bias column index is the absolute key pos
        bias = self.alibi(past_len + h.size(1), h.device, h.dtype)
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            out = self._run(layer, h, position_bias=bias, past_key_value=past,
                            use_cache=use_cache)
            if use_cache:
                h, p = out
                presents.append(p)
            else:
                h = out
        h = self.ln_f(h)
        logits = self.lm_head(h, self.word_embeddings.weight)
        if use_cache:
            return logits, presents
        return logits

    def set_activation_checkpoint(self, enabled=True):
        self.checkpoint_activations = enabled


class BloomLoss(nn.Module):
    def __init__(self):
        super().__init__()
        self.lm_loss = ParallelCrossEntropyLoss()

    def forward(self, logits, labels):
        return {"lm_loss": self.lm_loss(logits, labels).mean()}


class BloomForCausalLM(nn.Module):
    @configurable
    def __init__(self, cfg=None, **kwargs):
        super().__init__()
        self.bloom = BloomModel(cfg) if cfg is not None else BloomModel(**kwargs)
        self.loss_func = BloomLoss()
        self.hidden_size = self.bloom.hidden_size

    @classmethod
    def from_config(cls, cfg):
        return {"cfg": cfg}

    def forward(self, input_ids, labels=None, past_key_values=None,
                use_cache=False):
        if past_key_values is not None or use_cache:
            out = self.bloom(input_ids, past_key_values=past_key_values,
                             use_cache=use_cache)
            if use_cache:
                logits, presents = out
                return {"prediction_scores": logits, "past_key_values": presents}
            return {"prediction_scores": out}
        logits = self.bloom(input_ids)
        if labels is not None:
            return self.loss_func(logits, labels)
        return {"prediction_scores": logits}

    def set_activation_checkpoint(self, enabled=True):
        self.bloom.set_activation_checkpoint(enabled)

    # -- pipeline protocol --------------------------------------------------

    def pipeline_stage_batch_keys(self, is_first, is_last):
        keys = {"input_ids"}  # shape witness on middle stages
        if is_last:
            keys.add("labels")
        return keys

    def pipeline_units(self):
        bl = self.bloom

        def embed(h, b):
            return bl.word_embeddings_layernorm(bl.word_embeddings(b["input_ids"]))

        units = [(0, "embeddings", embed)]

        def layer_fn(lyr):
            def fn(h, b):
                bias = bl.alibi(h.size(1), h.device, h.dtype)
                return bl._run(lyr, h, position_bias=bias)
            return fn

        for i, layer in enumerate(bl.layers):
            units.append((i, f"layer_{i}", layer_fn(layer)))

        def head(h, b):
            logits = bl.lm_head(bl.ln_f(h), bl.word_embeddings.weight)
            if b.get("labels") is not None:
                return self.loss_func(logits, b["labels"])
            return {"prediction_scores": logits}

        units.append((-1, "head", head))
        return units

    def pipeline_stage_modules(self):
        bl = self.bloom
        m = {0: [bl.word_embeddings, bl.word_embeddings_layernorm]}
        for i, layer in enumerate(bl.layers):
            m.setdefault(i, []).append(layer)
        m.setdefault(-1, []).extend(
            [bl.ln_f, bl.lm_head, self.loss_func, bl.word_embeddings]
        )
        return m
