"""Llama model family (reference: projects/Llama/llama.py).

MI355X-native hot path: RMSNorm HIP kernel, RoPE HIP kernel on strided qkv
views, flash attention (head_dim 128 for 7B), fused SwiGLU MLP, untied
vocab-parallel output head + vocab-parallel CE.
"""

import math

import torch
from torch import nn
from torch.utils.checkpoint import checkpoint as act_checkpoint

from ..config import configurable
from ..layers import Linear1D, LMLogits, ParallelCrossEntropyLoss, RMSLayerNorm, VocabEmbedding
from ..ops.attention import flash_attention, flash_attention_available
from ..ops.rope import apply_rotary_pos_emb
from ..ops.softmax import fused_scale_mask_softmax
from ..ops.swiglu import swiglu
from ..utils import distributed as du
from .utils.weight_init import init_method_normal, scaled_init_method_normal

__all__ = ["LlamaModel", "LlamaForCausalLM"]


class LlamaAttention(nn.Module):
    """Self-attention with optional grouped-query KV heads (GQA — the
    Llama-2-70B / Qwen2 / ChatGLM2 sharing scheme; reference capability:
    projects/Qwen, projects/ChatGLM).  num_key_value_heads == num_heads is
    plain MHA with the fused qkv projection; fewer kv heads use a separate
    q projection + fused [k|v] pair, and the flash kernels map each query
    head to kv head h // (Hq/Hkv) in-kernel."""

    def __init__(self, hidden_size, num_heads, max_position_embeddings,
                 init_method, output_init_method, rope_theta=10000.0,
                 num_key_value_heads=None, sequence_parallel=False, *,
                 layer_idx=0):
        super().__init__()
        self.sequence_parallel = sequence_parallel
        self.hidden_size = hidden_size
        self.num_heads = num_heads
        self.head_dim = hidden_size // num_heads
        self.num_kv_heads = num_key_value_heads or num_heads
        assert num_heads % self.num_kv_heads == 0, (num_heads, self.num_kv_heads)
        dutil = du.get_dist_util()
        tp = dutil.tensor_parallel_size
        # the fused qkv weight is per-head interleaved; whole heads must land
        # on each TP rank for the contiguous column shard to stay coherent
        assert num_heads % tp == 0, (num_heads, tp)
        assert self.num_kv_heads % tp == 0, (self.num_kv_heads, tp)
        self.num_heads_local = num_heads // tp
        self.num_kv_local = self.num_kv_heads // tp
        self.max_pos = max_position_embeddings
        self.rope_theta = rope_theta
        self.layer_idx = layer_idx
        if self.num_kv_heads == num_heads:
            self.query_key_value = Linear1D(hidden_size, 3 * hidden_size,
                                            bias=False, parallel="col",
                                            init_method=init_method,
                                            sequence_parallel=sequence_parallel,
                                            layer_idx=layer_idx)
        else:
            self.q_proj = Linear1D(hidden_size, hidden_size, bias=False,
                                   parallel="col", init_method=init_method,
                                   sequence_parallel=sequence_parallel,
                                   layer_idx=layer_idx)
            self.kv_proj = Linear1D(
                hidden_size, 2 * self.num_kv_heads * self.head_dim, bias=False,
                parallel="col", init_method=init_method, fused_chunks=2,
                sequence_parallel=sequence_parallel, layer_idx=layer_idx)
        self.o_proj = Linear1D(hidden_size, hidden_size, bias=False, parallel="row",
                               init_method=output_init_method, skip_bias_add=True,
                               sequence_parallel=sequence_parallel,
                               layer_idx=layer_idx)
        self.scale = 1.0 / math.sqrt(self.head_dim)

    def _project(self, hidden_states):
        b, s, _ = hidden_states.shape
        if self.sequence_parallel:  # col projections gather the seq shards
            s = s * du.get_dist_util().tensor_parallel_size
        if self.num_kv_heads == self.num_heads:
            qkv = self.query_key_value(hidden_states)
            qkv5 = qkv.view(b, s, self.num_heads_local, 3, self.head_dim)
            return qkv5[..., 0, :], qkv5[..., 1, :], qkv5[..., 2, :]
        q = self.q_proj(hidden_states).view(b, s, self.num_heads_local,
                                            self.head_dim)
        kv = self.kv_proj(hidden_states).view(b, s, 2, self.num_kv_local,
                                              self.head_dim)
        return q, kv[:, :, 0], kv[:, :, 1]

    def forward(self, hidden_states, past_key_value=None, use_cache=False,
                residual=None, static_cache=None, position=None):
        if static_cache is not None:
            # hipGraph-capturable decode: RoPE at a device position tensor,
            # in-place KV insert into [b, kvh, MAX, hd] buffers, fused
            # flash_decode masked by device int32 kv_len (GQA in-kernel)
            from ..ops._ext import ext
            from ..ops.attention import flash_decode_attn
            from ..ops.rope import _tables

            ck, cv, kv32 = static_cache
            b, _, _ = hidden_states.shape
            q, k, v = self._project(hidden_states)
            # ONE kernel: RoPE(q) -> flash layout, RoPE(k)/copy(v) -> cache
            # row at the device position (replaces ~15 eager-shaped launches)
            cos_t, sin_t = _tables(self.max_pos, self.head_dim,
                                   self.rope_theta, hidden_states.device)
            qo = ext().rope_kv_insert(q, k, v, ck, cv, cos_t, sin_t, position, True)
            ctx = flash_decode_attn(qo, ck, cv, self.scale, kv_len=kv32)
            context = ctx.permute(0, 2, 1, 3).reshape(
                b, 1, self.num_heads_local * self.head_dim)
            out, _ = self.o_proj(context)
            return out  # RAW: the layer fuses residual + RMSNorm
        if self.sequence_parallel and (use_cache or past_key_value is not None):
            raise RuntimeError("sequence_parallel is training-only; build the "
                               "generation model with sequence_parallel=False")
        b, s, _ = hidden_states.shape
        if self.sequence_parallel:
            s = s * du.get_dist_util().tensor_parallel_size
        q, k, v = self._project(hidden_states)
        pos0 = past_key_value[0].shape[2] if past_key_value is not None else 0
        q = apply_rotary_pos_emb(q, self.max_pos, self.rope_theta, pos0)
        k = apply_rotary_pos_emb(k, self.max_pos, self.rope_theta, pos0)

        if (
            past_key_value is None
            and not use_cache
            and flash_attention_available(self.head_dim, q.dtype, q.device, s, s, None)
        ):
            o = flash_attention(q, k, v, self.scale, p_drop=0.0, causal=True,
                                training=self.training)
            context = o.reshape(b, s, self.num_heads_local * self.head_dim)
        else:
            # unfused path (decode / CPU): [b, nh, s, hs]
            group = self.num_heads_local // self.num_kv_local
            qh = q.permute(0, 2, 1, 3)
            kh = k.permute(0, 2, 1, 3)
            vh = v.permute(0, 2, 1, 3)
            if past_key_value is not None:
                pk, pv = past_key_value
                kh = torch.cat([pk, kh], dim=2)
                vh = torch.cat([pv, vh], dim=2)
            present = (kh, vh) if use_cache else None  # cache holds KV heads
            from ..ops.attention import (
                decode_attention_available,
                flash_decode_attn,
            )

            if past_key_value is not None and decode_attention_available(
                qh, self.head_dim
            ):
                # fused single-token decode (K16); GQA mapping in-kernel
                ctx = flash_decode_attn(qh.contiguous(), kh, vh, self.scale)
                context = ctx.permute(0, 2, 1, 3).reshape(
                    b, 1, self.num_heads_local * self.head_dim)
                out, _ = self.o_proj(context)
                out = out + residual if residual is not None else out
                return (out, present) if use_cache else out
            if group > 1:
                kh = kh.repeat_interleave(group, dim=1)
                vh = vh.repeat_interleave(group, dim=1)
            scores = torch.matmul(qh, kh.transpose(-1, -2))
            causal = past_key_value is None
            probs = fused_scale_mask_softmax(scores, scale=self.scale, causal=causal,
                                             training=self.training)
            ctx = torch.matmul(probs, vh)
            context = ctx.permute(0, 2, 1, 3).reshape(
                b, s, self.num_heads_local * self.head_dim
            )
            out, _ = self.o_proj(context)
            out = out + residual if residual is not None else out
            if use_cache:
                return out, present
            return out

        out, _ = self.o_proj(context)
        return out + residual if residual is not None else out


class LlamaMLP(nn.Module):
    """Gated MLP: fused [gate|up] col projection -> SwiGLU -> row projection."""

    def __init__(self, hidden_size, intermediate_size, init_method,
                 output_init_method, sequence_parallel=False, *, layer_idx=0):
        super().__init__()
        self.gate_up_proj = Linear1D(hidden_size, 2 * intermediate_size, bias=False,
                                     parallel="col", init_method=init_method,
                                     fused_chunks=2,
                                     sequence_parallel=sequence_parallel,
                                     layer_idx=layer_idx)
        self.down_proj = Linear1D(intermediate_size, hidden_size, bias=False,
                                  parallel="row", init_method=output_init_method,
                                  skip_bias_add=True,
                                  sequence_parallel=sequence_parallel,
                                  layer_idx=layer_idx)

    def forward(self, x, residual=None):
        gu = self.gate_up_proj(x)
        inter = swiglu(gu)
        out, _ = self.down_proj(inter)
        return out + residual if residual is not None else out


class LlamaDecoderLayer(nn.Module):
    def __init__(self, hidden_size, intermediate_size, num_heads,
                 max_position_embeddings, rms_norm_eps, init_method,
                 output_init_method, rope_theta=10000.0,
                 num_key_value_heads=None, sequence_parallel=False, *,
                 layer_idx=0):
        super().__init__()
        self.layer_idx = layer_idx
        self.input_layernorm = RMSLayerNorm(hidden_size, eps=rms_norm_eps,
                                            layer_idx=layer_idx)
        self.self_attn = LlamaAttention(hidden_size, num_heads,
                                        max_position_embeddings, init_method,
                                        output_init_method, rope_theta,
                                        num_key_value_heads,
                                        sequence_parallel=sequence_parallel,
                                        layer_idx=layer_idx)
        self.post_attention_layernorm = RMSLayerNorm(hidden_size, eps=rms_norm_eps,
                                                     layer_idx=layer_idx)
        self.mlp = LlamaMLP(hidden_size, intermediate_size, init_method,
                            output_init_method,
                            sequence_parallel=sequence_parallel,
                            layer_idx=layer_idx)
        if sequence_parallel:
            # RMSNorms run on seq shards: grads are partial over local tokens
            for ln in (self.input_layernorm, self.post_attention_layernorm):
                for p in ln.parameters():
                    p.sequence_parallel_grad = True

    def forward(self, hidden_states, past_key_value=None, use_cache=False,
                static_cache=None, position=None):
        if static_cache is not None:
            # residual + post-attention RMSNorm fused into one kernel
            from ..ops._ext import ext

            ln1 = self.input_layernorm(hidden_states)
            a = self.self_attn(ln1, static_cache=static_cache,
                               position=position)
            ln = self.post_attention_layernorm
            if a.shape[-1] <= 2048:
                h, ln2 = ext().res_norm_fwd(a, None, hidden_states, ln.weight,
                                            None, ln.eps, True)
            else:  # wide rows (e.g. 7B h4096): separate add + RMSNorm
                h = a + hidden_states
                ln2 = ln(h)
            return self.mlp(ln2, residual=h)
        ln1 = self.input_layernorm(hidden_states)
        attn_out = self.self_attn(ln1, past_key_value=past_key_value,
                                  use_cache=use_cache, residual=hidden_states)
        if use_cache:
            attn_out, present = attn_out
        h = attn_out
        ln2 = self.post_attention_layernorm(h)
        out = self.mlp(ln2, residual=h)
        if use_cache:
            return out, present
        return out


class LlamaModel(nn.Module):
    @configurable
    def __init__(
        self,
        hidden_layers,
        vocab_size,
        hidden_size,
        intermediate_size,
        num_attention_heads,
        max_position_embeddings=2048,
        rms_norm_eps=1e-5,
        initializer_range=0.02,
        use_scaled_init_for_output_weights=False,
        tie_word_embeddings=False,
        rope_theta=10000.0,
        num_key_value_heads=None,
        sequence_parallel=False,
        amp_enabled=False,
    ):
        super().__init__()
        self.sequence_parallel = sequence_parallel
        init_method = init_method_normal(initializer_range)
        output_init_method = (
            scaled_init_method_normal(initializer_range, hidden_layers)
            if use_scaled_init_for_output_weights
            else init_method
        )
        self.embed_tokens = VocabEmbedding(vocab_size, hidden_size,
                                           init_method=init_method, layer_idx=0)
        self.layers = nn.ModuleList(
            [
                LlamaDecoderLayer(
                    hidden_size, intermediate_size, num_attention_heads,
                    max_position_embeddings, rms_norm_eps, init_method,
                    output_init_method, rope_theta, num_key_value_heads,
                    sequence_parallel=sequence_parallel,
                    layer_idx=i,
                )
                for i in range(hidden_layers)
            ]
        )
        self.norm = RMSLayerNorm(hidden_size, eps=rms_norm_eps, layer_idx=-1)
        if sequence_parallel:
            for p in self.norm.parameters():
                p.sequence_parallel_grad = True
        self.hidden_layers = hidden_layers
        self.checkpoint_activations = False

    @classmethod
    def from_config(cls, cfg):
        return {
            "hidden_layers": cfg.hidden_layers,
            "vocab_size": cfg.vocab_size,
            "hidden_size": cfg.hidden_size,
            "intermediate_size": cfg.intermediate_size,
            "num_attention_heads": cfg.num_attention_heads,
            "max_position_embeddings": cfg.get("max_position_embeddings", 2048),
            "rms_norm_eps": cfg.get("rms_norm_eps", 1e-5),
            "initializer_range": cfg.get("initializer_range", 0.02),
            "use_scaled_init_for_output_weights": cfg.get(
                "use_scaled_init_for_output_weights", False
            ),
            "tie_word_embeddings": cfg.get("tie_word_embeddings", False),
            "rope_theta": cfg.get("rope_theta", 10000.0),
            "num_key_value_heads": cfg.get("num_key_value_heads", None),
            "sequence_parallel": cfg.get("sequence_parallel", False),
            "amp_enabled": cfg.get("amp_enabled", False),
        }

    def _run_layer(self, layer, h, past=None, use_cache=False):
        if self.checkpoint_activations and self.training and not use_cache:
            return act_checkpoint(layer, h, use_reentrant=False)
        return layer(h, past_key_value=past, use_cache=use_cache)

    def forward(self, input_ids, past_key_values=None, use_cache=False,
                static_caches=None, position=None):
        if static_caches is not None:
            h = self.embed_tokens(input_ids)
            for layer, sc in zip(self.layers, static_caches):
                h = layer(h, static_cache=sc, position=position)
            return self.norm(h)
        h = self.embed_tokens(input_ids)
        if self.sequence_parallel and not use_cache and past_key_values is None:
            from ..parallel.comm import scatter_to_sequence_parallel_region

            assert input_ids.shape[1] % du.get_dist_util().tensor_parallel_size == 0
            h = scatter_to_sequence_parallel_region(h)
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            h = self._run_layer(layer, h, past, use_cache)
            if use_cache:
                h, p = h
                presents.append(p)
        h = self.norm(h)
        if use_cache:
            return h, presents
        return h

    def set_activation_checkpoint(self, enabled=True):
        self.checkpoint_activations = enabled


class _LMHeadWeight(nn.Module):
    """Vocab-sharded untied output-projection weight (module-wrapped so the
    pipeline prune places it on the last stage only)."""

    def __init__(self, vocab_size, hidden_size, init_method):
        super().__init__()
        dutil = du.get_dist_util()
        tp = dutil.tensor_parallel_size
        self.weight = nn.Parameter(torch.empty(vocab_size // tp, hidden_size))
        self.weight.tensor_parallel = True
        self.weight.tp_shard_dim = 0
        from ..layers.linear import init_tp_shard_

        init_tp_shard_(self.weight, (vocab_size, hidden_size), init_method, 0)


class LlamaLoss(nn.Module):
    def __init__(self):
        super().__init__()
        self.lm_loss = ParallelCrossEntropyLoss()

    def forward(self, logits, labels):
        return {"lm_loss": self.lm_loss(logits, labels).mean()}


class LlamaForCausalLM(nn.Module):
    @configurable
    def __init__(self, cfg=None, **kwargs):
        super().__init__()
        self.model = LlamaModel(cfg) if cfg is not None else LlamaModel(**kwargs)
        vocab = cfg.vocab_size if cfg is not None else kwargs["vocab_size"]
        hidden = cfg.hidden_size if cfg is not None else kwargs["hidden_size"]
        tie = (cfg.get("tie_word_embeddings", False) if cfg is not None
               else kwargs.get("tie_word_embeddings", False))
        self.tie_word_embeddings = tie
        if not tie:
            init_range = (cfg.get("initializer_range", 0.02) if cfg is not None
                          else kwargs.get("initializer_range", 0.02))
            self.lm_head = _LMHeadWeight(vocab, hidden,
                                         init_method_normal(init_range))
        sp = (cfg.get("sequence_parallel", False) if cfg is not None
              else kwargs.get("sequence_parallel", False))
        self.lm_logits = LMLogits(vocab, bias=False, sequence_parallel=sp,
                                  layer_idx=-1)
        self.loss_func = LlamaLoss()

    @classmethod
    def from_config(cls, cfg):
        return {"cfg": cfg}

    def _head_weight(self):
        return (self.model.embed_tokens.weight if self.tie_word_embeddings
                else self.lm_head.weight)

    def forward(self, input_ids, labels=None, past_key_values=None, use_cache=False,
                static_caches=None, position=None):
        if static_caches is not None:
            h = self.model(input_ids, static_caches=static_caches,
                           position=position)
            return self.lm_logits(h, self._head_weight())
        h = self.model(input_ids, past_key_values=past_key_values, use_cache=use_cache)
        if use_cache:
            h, presents = h
        logits = self.lm_logits(h, self._head_weight())
        if labels is not None:
            return self.loss_func(logits, labels)
        if use_cache:
            return {"prediction_scores": logits, "past_key_values": presents}
        return {"prediction_scores": logits}

    def set_activation_checkpoint(self, enabled=True):
        self.model.set_activation_checkpoint(enabled)

    # -- pipeline protocol --------------------------------------------------

    def pipeline_stage_batch_keys(self, is_first, is_last):
        keys = {"input_ids"}
        if is_last:
            keys.add("labels")
        return keys

    def pipeline_units(self):
        units = [(0, "embed_tokens", lambda h, b: self.model.embed_tokens(b["input_ids"]))]
        for i, layer in enumerate(self.model.layers):
            units.append(
                (i, f"layer_{i}",
                 (lambda lyr: lambda h, b: self.model._run_layer(lyr, h))(layer))
            )

        def head(h, b):
            h = self.model.norm(h)
            logits = self.lm_logits(h, self._head_weight())
            if b.get("labels") is not None:
                return self.loss_func(logits, b["labels"])
            return {"prediction_scores": logits}

        units.append((-1, "head", head))
        return units

    def pipeline_stage_modules(self):
        m = {0: [self.model.embed_tokens]}
        for i, layer in enumerate(self.model.layers):
            m.setdefault(i, []).append(layer)
        last = [self.model.norm, self.lm_logits, self.loss_func]
        if self.tie_word_embeddings:
            last.append(self.model.embed_tokens)
        else:
            last.append(self.lm_head)
        m.setdefault(-1, []).extend(last)
        return m
