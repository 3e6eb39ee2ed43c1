"""GPT-2 model family (reference: libai/models/gpt_model.py:118-401).

MI355X-native differences from the reference:
  * no materialized causal-mask tensor — causality is an in-kernel predicate
    of the fused softmax (replaces CasualMask, gpt_model.py:42-51 / K14)
  * pipeline placement is metadata (``layer_idx`` on each module) consumed by
    the explicit 1F1B engine via ``pipeline_units()`` instead of OneFlow
    graph stage ids (gpt_model.py:359-401)
  * activation checkpointing wraps each TransformerLayer with
    torch.utils.checkpoint (non-reentrant; philox dropout seeds ride the
    preserved CPU RNG state).
"""

import torch
from torch import nn
from torch.utils.checkpoint import checkpoint as act_checkpoint

from ..config import configurable
from ..utils import distributed as du
from ..layers import (
    AttnMaskType,
    Embedding,
    LayerNorm,
    LMLogits,
    ParallelCrossEntropyLoss,
    TransformerLayer,
    VocabEmbedding,
)
from .utils.weight_init import init_method_normal, scaled_init_method_normal

__all__ = ["GPTEmbedding", "GPTModel", "GPTLoss", "GPTForPreTraining"]


class GPTEmbedding(nn.Module):
    def __init__(self, vocab_size, hidden_size, max_seq_length,
                 init_method, embedding_dropout_prob=0.0, *, layer_idx=0):
        super().__init__()
        self.layer_idx = layer_idx
        self.token_embeddings = VocabEmbedding(
            vocab_size, hidden_size, init_method=init_method, layer_idx=layer_idx
        )
        self.position_embeddings = Embedding(
            max_seq_length, hidden_size, init_method=init_method, layer_idx=layer_idx
        )
        self.dropout = nn.Dropout(embedding_dropout_prob)
        self.register_buffer(
            "position_ids", torch.arange(max_seq_length).unsqueeze(0), persistent=False
        )

    def forward(self, input_ids, past_length=0):
        seq_len = input_ids.size(1)
        if torch.is_tensor(past_length):
            # device-tensor position (hipGraph-captured decode: no host ints)
            pos_ids = past_length.view(-1, 1) + torch.arange(
                seq_len, device=input_ids.device).view(1, -1)
        else:
            pos_ids = self.position_ids[:, past_length : past_length + seq_len]
        embeds = self.token_embeddings(input_ids) + self.position_embeddings(pos_ids)
        return self.dropout(embeds)


class Transformer(nn.Module):
    def __init__(self, num_layers, hidden_size, ffn_hidden_size, num_attention_heads,
                 attention_dropout_prob, output_dropout_prob, layernorm_epsilon,
                 init_method, output_layer_init_method, bias_gelu_fusion,
                 bias_dropout_fusion, scale_mask_softmax_fusion,
                 apply_query_key_layer_scaling, apply_residual_post_layernorm,
                 sequence_parallel=False, moe_num_experts=0, moe_top_k=2):
        super().__init__()
        self.num_layers = num_layers
        self.checkpoint_activations = False
        self.layers = nn.ModuleList(
            [
                TransformerLayer(
                    hidden_size, ffn_hidden_size, num_attention_heads,
                    attention_dropout_prob=attention_dropout_prob,
                    output_dropout_prob=output_dropout_prob,
                    layernorm_epsilon=layernorm_epsilon,
                    init_method=init_method,
                    output_layer_init_method=output_layer_init_method,
                    bias_gelu_fusion=bias_gelu_fusion,
                    bias_dropout_fusion=bias_dropout_fusion,
                    scale_mask_softmax_fusion=scale_mask_softmax_fusion,
                    apply_query_key_layer_scaling=apply_query_key_layer_scaling,
                    apply_residual_post_layernorm=apply_residual_post_layernorm,
                    attn_mask_type=AttnMaskType.causal,
                    sequence_parallel=sequence_parallel,
                    mlp_type="moe" if moe_num_experts else "dense",
                    moe_num_experts=moe_num_experts or 8,
                    moe_top_k=moe_top_k,
                    layer_idx=i,
                )
                for i in range(num_layers)
            ]
        )
        self.layernorm_f = LayerNorm(hidden_size, eps=layernorm_epsilon, layer_idx=-1)

    def _run_layer(self, layer, hidden_states, past_key_value=None, use_cache=False):
        if self.checkpoint_activations and self.training and not use_cache:
            return act_checkpoint(layer, hidden_states, use_reentrant=False)
        return layer(hidden_states, past_key_value=past_key_value, use_cache=use_cache)

    def forward(self, hidden_states, past_key_values=None, use_cache=False,
                static_caches=None, position=None):
        if static_caches is not None:
            for layer, sc in zip(self.layers, static_caches):
                hidden_states = layer(hidden_states, static_cache=sc,
                                      position=position)
            return self.layernorm_f(hidden_states)
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            hidden_states = self._run_layer(layer, hidden_states, past, use_cache)
            if use_cache:
                hidden_states, present = hidden_states
                presents.append(present)
        out = self.layernorm_f(hidden_states)
        if use_cache:
            return out, presents
        return out


class GPTModel(nn.Module):
    @configurable
    def __init__(
        self,
        hidden_layers,
        vocab_size,
        hidden_size,
        ffn_hidden_size,
        num_attention_heads,
        max_seq_length=1024,
        embedding_dropout_prob=0.0,
        attention_dropout_prob=0.0,
        output_dropout_prob=0.0,
        layernorm_epsilon=1e-5,
        initializer_range=0.02,
        use_scaled_init_for_output_weights=True,
        bias_gelu_fusion=True,
        bias_dropout_fusion=True,
        scale_mask_softmax_fusion=True,
        apply_query_key_layer_scaling=False,
        apply_residual_post_layernorm=False,
        sequence_parallel=False,
        moe_num_experts=0,
        moe_top_k=2,
        amp_enabled=False,
    ):
        super().__init__()
        self.sequence_parallel = sequence_parallel
        init_method = init_method_normal(initializer_range)
        output_layer_init_method = (
            scaled_init_method_normal(initializer_range, hidden_layers)
            if use_scaled_init_for_output_weights
            else init_method
        )
        self.embeddings = GPTEmbedding(
            vocab_size, hidden_size, max_seq_length,
            init_method=init_method,
            embedding_dropout_prob=embedding_dropout_prob,
            layer_idx=0,
        )
        self.transformer = Transformer(
            hidden_layers, hidden_size, ffn_hidden_size, num_attention_heads,
            attention_dropout_prob, output_dropout_prob, layernorm_epsilon,
            init_method, output_layer_init_method, bias_gelu_fusion,
            bias_dropout_fusion, scale_mask_softmax_fusion,
            apply_query_key_layer_scaling, apply_residual_post_layernorm,
            sequence_parallel=sequence_parallel,
            moe_num_experts=moe_num_experts, moe_top_k=moe_top_k,
        )
        self.lm_head = LMLogits(vocab_size, bias=False,
                                sequence_parallel=sequence_parallel, layer_idx=-1)
        if sequence_parallel:
            for p in self.transformer.layernorm_f.parameters():
                p.sequence_parallel_grad = True  # runs on the seq shard
        self.hidden_layers = hidden_layers

    @classmethod
    def from_config(cls, cfg):
        return {
            "hidden_layers": cfg.hidden_layers,
            "vocab_size": cfg.vocab_size,
            "hidden_size": cfg.hidden_size,
            "ffn_hidden_size": cfg.ffn_hidden_size,
            "num_attention_heads": cfg.num_attention_heads,
            "max_seq_length": cfg.get("max_seq_length", 1024),
            "embedding_dropout_prob": cfg.get("embedding_dropout_prob", 0.0),
            "attention_dropout_prob": cfg.get("attention_dropout_prob", 0.0),
            "output_dropout_prob": cfg.get("output_dropout_prob", 0.0),
            "layernorm_epsilon": cfg.get("layernorm_epsilon", 1e-5),
            "initializer_range": cfg.get("initializer_range", 0.02),
            "use_scaled_init_for_output_weights": cfg.get(
                "use_scaled_init_for_output_weights", True
            ),
            "bias_gelu_fusion": cfg.get("bias_gelu_fusion", True),
            "bias_dropout_fusion": cfg.get("bias_dropout_fusion", True),
            "scale_mask_softmax_fusion": cfg.get("scale_mask_softmax_fusion", True),
            "apply_query_key_layer_scaling": cfg.get(
                "apply_query_key_layer_scaling", False
            ),
            "apply_residual_post_layernorm": cfg.get(
                "apply_residual_post_layernorm", False
            ),
            "sequence_parallel": cfg.get("sequence_parallel", False),
            "moe_num_experts": cfg.get("moe_num_experts", 0),
            "moe_top_k": cfg.get("moe_top_k", 2),
            "amp_enabled": cfg.get("amp_enabled", False),
        }

    def forward(self, input_ids, past_key_values=None, use_cache=False,
                static_caches=None, position=None):
        if static_caches is not None:
            # hipGraph-captured decode: all shapes static, position/kv_len are
            # DEVICE tensors, KV written in place into preallocated buffers
            h = self.embeddings(input_ids, position)
            h = self.transformer(h, static_caches=static_caches,
                                 position=position)
            return self.lm_head(h, self.embeddings.token_embeddings.weight)
        past_length = (
            past_key_values[0][0].shape[2] if past_key_values is not None else 0
        )
        h = self.embeddings(input_ids, past_length)
        if self.sequence_parallel and (use_cache or past_key_values is not None):
            raise RuntimeError(
                "sequence_parallel is a training-time sharding; build the "
                "generation model with sequence_parallel=False (checkpoints "
                "are topology-independent and load either way)"
            )
        if self.sequence_parallel and not use_cache and past_key_values is None:
            # SP region entry: [b, s, h] -> this rank's [b, s/tp, h] shard
            from ..parallel.comm import scatter_to_sequence_parallel_region

            assert input_ids.shape[1] % du.get_dist_util().tensor_parallel_size == 0
            h = scatter_to_sequence_parallel_region(h)
        h = self.transformer(h, past_key_values=past_key_values, use_cache=use_cache)
        if use_cache:
            h, presents = h
        logits = self.lm_head(h, self.embeddings.token_embeddings.weight)
        if use_cache:
            return logits, presents
        return logits

    def set_activation_checkpoint(self, enabled=True):
        self.transformer.checkpoint_activations = enabled


class GPTLoss(nn.Module):
    def __init__(self):
        super().__init__()
        self.lm_loss = ParallelCrossEntropyLoss()

    def forward(self, logits, lm_labels):
        return {"lm_loss": self.lm_loss(logits, lm_labels).mean()}


class GPTForPreTraining(nn.Module):
    @configurable
    def __init__(self, cfg=None, **kwargs):
        super().__init__()
        self.GPT_model = GPTModel(cfg) if cfg is not None else GPTModel(**kwargs)
        self.loss_func = GPTLoss()

    @classmethod
    def from_config(cls, cfg):
        return {"cfg": cfg}

    def forward(self, input_ids, labels=None, past_key_values=None,
                use_cache=False):
        if past_key_values is not None or use_cache:
            # incremental-decode path for the generation pipelines
            out = self.GPT_model(input_ids, past_key_values=past_key_values,
                                 use_cache=use_cache)
            if use_cache:
                logits, presents = out
                return {"prediction_scores": logits, "past_key_values": presents}
            return {"prediction_scores": out}
        logits = self.GPT_model(input_ids)
        if labels is not None:
            out = self.loss_func(logits, labels)
            aux = [m.last_aux_loss for m in self.GPT_model.modules()
                   if getattr(m, "last_aux_loss", None) is not None]
            if aux:
                out["moe_aux_loss"] = torch.stack(aux).sum()
            return out
        return {"prediction_scores": logits}

    def set_activation_checkpoint(self, enabled=True):
        self.GPT_model.set_activation_checkpoint(enabled)

    def pipeline_stage_batch_keys(self, is_first, is_last):
        """Batch tensors this pipeline stage actually consumes (middle
        stages read nothing; the activations arrive via P2P)."""
        keys = set()
        if is_first:
            keys.add("input_ids")
        if is_last:
            keys.add("labels")
        return keys

    # -- pipeline protocol --------------------------------------------------

    def pipeline_units(self):
        """Ordered (layer_idx, name, fn(hidden, batch)) units for the 1F1B
        engine; first unit ignores ``hidden``, last returns the loss dict."""
        assert not getattr(self.GPT_model, "sequence_parallel", False), (
            "sequence parallelism is not composed with pipeline parallelism"
        )
        units = [
            (0, "embeddings", lambda h, b: self.GPT_model.embeddings(b["input_ids"]))
        ]
        for i, layer in enumerate(self.GPT_model.transformer.layers):
            units.append(
                (
                    i,
                    f"layer_{i}",
                    (lambda lyr: lambda h, b: self.GPT_model.transformer._run_layer(lyr, h))(
                        layer
                    ),
                )
            )

        def head(h, b):
            h = self.GPT_model.transformer.layernorm_f(h)
            logits = self.GPT_model.lm_head(
                h, self.GPT_model.embeddings.token_embeddings.weight
            )
            if b.get("labels") is not None:
                return self.loss_func(logits, b["labels"])
            return {"prediction_scores": logits}

        units.append((-1, "head", head))
        return units

    def pipeline_stage_modules(self):
        """layer_idx -> [modules] map used to prune non-local stages.

        The tied token embedding is listed on BOTH the first and last stage
        (the lm_head reads it), which makes the pipeline engine keep a copy on
        each and all-reduce its grad over the tied-parameter group — the
        explicit version of the reference moving the embedding weight to the
        last stage (libai/layers/lm_logits.py:44).
        """
        m = {0: [self.GPT_model.embeddings]}
        for i, layer in enumerate(self.GPT_model.transformer.layers):
            m.setdefault(i, []).append(layer)
        m.setdefault(-1, []).extend(
            [self.GPT_model.transformer.layernorm_f, self.GPT_model.lm_head,
             self.loss_func, self.GPT_model.embeddings.token_embeddings]
        )
        return m
