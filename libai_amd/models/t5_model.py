"""T5 encoder-decoder model (reference: libai/models/t5_model.py:335-518).

Megatron-style T5: shared vocab-parallel embedding, encoder stack (padding
mask), decoder stack with causal self-attention + cross-attention to the
encoder output, tied LM logits, masked-LM loss.  KV caches cover both the
decoder self-attention and the cross-attention (reference t5_model.py:335-346).
"""

import torch
from torch import nn
from torch.utils.checkpoint import checkpoint as act_checkpoint

from ..config import configurable
from ..layers import (
    AttnMaskType,
    Embedding,
    LayerNorm,
    LMLogits,
    ParallelCrossEntropyLoss,
    TransformerLayer,
    VocabEmbedding,
)
from .bert_model import extended_attn_mask
from .utils.weight_init import init_method_normal, scaled_init_method_normal

__all__ = ["T5Model", "T5ForPreTraining", "T5Loss"]


def cross_attn_mask(dec_mask, enc_mask):
    """[b, sq] x [b, sk] visibilities -> [b, sq, sk] uint8 (1=masked)."""
    if dec_mask is None or enc_mask is None:
        return None
    vis = dec_mask.to(torch.uint8).unsqueeze(2) * enc_mask.to(torch.uint8).unsqueeze(1)
    return (1 - vis).to(torch.uint8)


class T5Embedding(nn.Module):
    def __init__(self, vocab_size, hidden_size, max_position_embeddings,
                 embedding_dropout_prob, init_method, *, layer_idx=0,
                 use_position_embeddings=True):
        super().__init__()
        self.word_embeddings = VocabEmbedding(vocab_size, hidden_size,
                                              init_method=init_method,
                                              layer_idx=layer_idx)
        # T5 with relative attention biases has NO absolute positions
        self.position_embeddings = (
            Embedding(max_position_embeddings, hidden_size,
                      init_method=init_method, layer_idx=layer_idx)
            if use_position_embeddings else None
        )
        self.dropout = nn.Dropout(embedding_dropout_prob)
        self.register_buffer(
            "position_ids", torch.arange(max_position_embeddings).unsqueeze(0),
            persistent=False,
        )

    def forward(self, input_ids, past_length=0):
        emb = self.word_embeddings(input_ids)
        if self.position_embeddings is not None:
            s = input_ids.size(1)
            pos = self.position_ids[:, past_length : past_length + s]
            emb = emb + self.position_embeddings(pos)
        return self.dropout(emb)


class T5Model(nn.Module):
    @configurable
    def __init__(
        self,
        vocab_size,
        hidden_size,
        hidden_layers,
        num_attention_heads,
        intermediate_size,
        hidden_dropout_prob=0.1,
        attention_probs_dropout_prob=0.1,
        max_position_embeddings=512,
        embedding_dropout_prob=0.1,
        initializer_range=0.02,
        layernorm_eps=1e-5,
        bias_gelu_fusion=True,
        bias_dropout_fusion=True,
        scale_mask_softmax_fusion=True,
        apply_query_key_layer_scaling=False,
        apply_residual_post_layernorm=False,
        amp_enabled=False,
        mlp_type="dense",
        activation="gelu",
        relative_attention=False,
        relative_attention_num_buckets=32,
        relative_attention_max_distance=128,
    ):
        super().__init__()
        init_method = init_method_normal(initializer_range)
        scaled_init = scaled_init_method_normal(initializer_range, hidden_layers)
        self.hidden_layers = hidden_layers
        self.hidden_size = hidden_size

        self.embedding = T5Embedding(
            vocab_size, hidden_size, max_position_embeddings,
            embedding_dropout_prob, init_method, layer_idx=0,
            use_position_embeddings=not relative_attention,
        )
        # MT5/T5.1.1 relative-position bias (reference
        # projects/MT5/layers/attention_layer.py:118-123): one learned
        # bucketed bias per stack, shared by every layer in it
        self.enc_rel_bias = self.dec_rel_bias = None
        if relative_attention:
            from ..layers.position_bias import T5RelativePositionBias

            self.enc_rel_bias = T5RelativePositionBias(
                num_attention_heads, relative_attention_num_buckets,
                relative_attention_max_distance, bidirectional=True,
                init_method=init_method, layer_idx=0,
            )
            self.dec_rel_bias = T5RelativePositionBias(
                num_attention_heads, relative_attention_num_buckets,
                relative_attention_max_distance, bidirectional=False,
                init_method=init_method, layer_idx=hidden_layers,
            )

        def make_layer(i, is_decoder):
            return TransformerLayer(
                hidden_size, intermediate_size, num_attention_heads,
                is_decoder=is_decoder,
                attention_dropout_prob=attention_probs_dropout_prob,
                output_dropout_prob=hidden_dropout_prob,
                layernorm_epsilon=layernorm_eps,
                init_method=init_method,
                output_layer_init_method=scaled_init,
                bias_gelu_fusion=bias_gelu_fusion,
                bias_dropout_fusion=bias_dropout_fusion,
                scale_mask_softmax_fusion=scale_mask_softmax_fusion,
                apply_query_key_layer_scaling=apply_query_key_layer_scaling,
                apply_residual_post_layernorm=apply_residual_post_layernorm,
                attn_mask_type=(
                    AttnMaskType.causal if is_decoder else AttnMaskType.padding
                ),
                mlp_type=mlp_type,
                activation=activation,
                layer_idx=i,
            )

        # layer_idx spans encoder (0..n-1) then decoder (n..2n-1) for PP
        self.encoder_layers = nn.ModuleList(
            [make_layer(i, False) for i in range(hidden_layers)]
        )
        self.encoder_final_layernorm = LayerNorm(hidden_size, eps=layernorm_eps,
                                                 layer_idx=hidden_layers - 1)
        self.decoder_layers = nn.ModuleList(
            [make_layer(hidden_layers + i, True) for i in range(hidden_layers)]
        )
        self.decoder_final_layernorm = LayerNorm(hidden_size, eps=layernorm_eps,
                                                 layer_idx=-1)
        self.lm_head = LMLogits(vocab_size, bias=True, layer_idx=-1)
        self.checkpoint_activations = False

    @classmethod
    def from_config(cls, cfg):
        return {
            "vocab_size": cfg.vocab_size,
            "hidden_size": cfg.hidden_size,
            "hidden_layers": cfg.hidden_layers,
            "num_attention_heads": cfg.num_attention_heads,
            "intermediate_size": cfg.intermediate_size,
            "hidden_dropout_prob": cfg.get("hidden_dropout_prob", 0.1),
            "attention_probs_dropout_prob": cfg.get("attention_probs_dropout_prob", 0.1),
            "max_position_embeddings": cfg.get("max_position_embeddings", 512),
            "embedding_dropout_prob": cfg.get("embedding_dropout_prob", 0.1),
            "initializer_range": cfg.get("initializer_range", 0.02),
            "layernorm_eps": cfg.get("layernorm_eps", 1e-5),
            "bias_gelu_fusion": cfg.get("bias_gelu_fusion", True),
            "bias_dropout_fusion": cfg.get("bias_dropout_fusion", True),
            "scale_mask_softmax_fusion": cfg.get("scale_mask_softmax_fusion", True),
            "apply_query_key_layer_scaling": cfg.get("apply_query_key_layer_scaling", False),
            "apply_residual_post_layernorm": cfg.get("apply_residual_post_layernorm", False),
            "amp_enabled": cfg.get("amp_enabled", False),
            "mlp_type": cfg.get("mlp_type", "dense"),
            "activation": cfg.get("activation", "gelu"),
            "relative_attention": cfg.get("relative_attention", False),
            "relative_attention_num_buckets": cfg.get(
                "relative_attention_num_buckets", 32),
            "relative_attention_max_distance": cfg.get(
                "relative_attention_max_distance", 128),
        }

    def _run(self, layer, *args, **kw):
        if self.checkpoint_activations and self.training:
            return act_checkpoint(layer, *args, use_reentrant=False, **kw)
        return layer(*args, **kw)

    def encode(self, encoder_input_ids, encoder_attn_mask=None):
        mask = extended_attn_mask(encoder_attn_mask)
        h = self.embedding(encoder_input_ids)
        s = h.size(1)
        bias = self.enc_rel_bias(s, s, h.device) if self.enc_rel_bias else None
        for layer in self.encoder_layers:
            h = self._run(layer, h, mask, position_bias=bias)
        return self.encoder_final_layernorm(h)

    def forward(
        self,
        encoder_input_ids,
        decoder_input_ids,
        encoder_attn_mask=None,
        decoder_attn_mask=None,
        encoder_decoder_attn_mask=None,
        encoder_states=None,
        past_key_values=None,
        use_cache=False,
    ):
        if encoder_states is None:
            encoder_states = self.encode(encoder_input_ids, encoder_attn_mask)
        if encoder_decoder_attn_mask is None and encoder_attn_mask is not None:
            dm = decoder_attn_mask if decoder_attn_mask is not None else torch.ones_like(
                decoder_input_ids
            )
            encoder_decoder_attn_mask = cross_attn_mask(dm, encoder_attn_mask)

        past_len = (
            past_key_values[0][0][0].shape[2] if past_key_values is not None else 0
        )
        h = self.embedding(decoder_input_ids, past_len)
        dec_bias = None
        if self.dec_rel_bias is not None:
            sk = h.size(1) + past_len
            dec_bias = self.dec_rel_bias(h.size(1), sk, h.device) if past_len == 0 \
                else self.dec_rel_bias(sk, sk, h.device)[:, :, past_len:, :]
        presents = [] if use_cache else None
        for i, layer in enumerate(self.decoder_layers):
            past = past_key_values[i] if past_key_values is not None else None
            out = self._run(
                layer, h,
                attention_mask=None,  # causal handled in-kernel
                encoder_states=encoder_states,
                encoder_attention_mask=encoder_decoder_attn_mask,
                past_key_value=past,
                use_cache=use_cache,
                position_bias=dec_bias,
            )
            if use_cache:
                h, p = out
                presents.append(p)
            else:
                h = out
        h = self.decoder_final_layernorm(h)
        logits = self.lm_head(h, self.embedding.word_embeddings.weight)
        if use_cache:
            return logits, encoder_states, presents
        return logits

    def set_activation_checkpoint(self, enabled=True):
        self.checkpoint_activations = enabled


class T5Loss(nn.Module):
    def __init__(self):
        super().__init__()
        self.lm_loss = ParallelCrossEntropyLoss()

    def forward(self, logits, lm_labels, loss_mask):
        loss = self.lm_loss(logits, lm_labels.clamp(min=0))
        lm = loss.view(-1) * loss_mask.float().view(-1)
        return {"masked_lm_loss": lm.sum() / loss_mask.float().sum().clamp(min=1.0)}


class T5ForPreTraining(nn.Module):
    @configurable
    def __init__(self, cfg=None, **kwargs):
        super().__init__()
        self.t5_model = T5Model(cfg) if cfg is not None else T5Model(**kwargs)
        self.loss_func = T5Loss()

    @classmethod
    def from_config(cls, cfg):
        return {"cfg": cfg}

    def forward(self, encoder_input_ids, decoder_input_ids, encoder_attn_mask=None,
                decoder_attn_mask=None, encoder_decoder_attn_mask=None,
                lm_labels=None, loss_mask=None):
        logits = self.t5_model(
            encoder_input_ids, decoder_input_ids, encoder_attn_mask,
            decoder_attn_mask, encoder_decoder_attn_mask,
        )
        if lm_labels is not None:
            return self.loss_func(logits, lm_labels, loss_mask)
        return {"prediction_scores": logits}

    def set_activation_checkpoint(self, enabled=True):
        self.t5_model.set_activation_checkpoint(enabled)

    # -- pipeline-parallel protocol -----------------------------------------
    # layer_idx space: encoder 0..n-1 (embedding at 0, enc final LN at n-1),
    # decoder n..2n-1 (decoder embedding at n), head at -1 — so
    # cfg.train.dist.pipeline_num_layers must be 2*hidden_layers (reference
    # t5_model.py:450+ assigns stage ids the same way via OneFlow placements).
    # The boundary state is (enc_out,) inside/after the encoder and
    # (enc_out, dec_hidden) between decoder stages — the multi-tensor case
    # the tuple-boundary scheduler exists for.

    @staticmethod
    def _enc_mask(b):
        if "_enc_mask" not in b:
            b["_enc_mask"] = extended_attn_mask(b.get("encoder_attn_mask"))
        return b["_enc_mask"]

    @staticmethod
    def _cross_mask(b):
        if "_cross_mask" not in b:
            enc = b.get("encoder_attn_mask")
            if enc is None:
                b["_cross_mask"] = None
            else:
                dm = b.get("decoder_attn_mask")
                if dm is None:
                    dm = torch.ones_like(b["decoder_input_ids"])
                b["_cross_mask"] = cross_attn_mask(dm, enc)
        return b["_cross_mask"]

    def pipeline_stage_batch_keys(self, is_first, is_last):
        keys = {"encoder_input_ids", "decoder_input_ids",
                "encoder_attn_mask", "decoder_attn_mask"}
        if is_last:
            keys.update({"lm_labels", "loss_mask"})
        return keys

    def pipeline_units(self):
        t5 = self.t5_model
        n = t5.hidden_layers
        units = [
            (0, "enc_embedding",
             lambda h, b: t5.embedding(b["encoder_input_ids"]))
        ]
        def enc_bias(b, s, device):
            if t5.enc_rel_bias is None:
                return None
            # one bias tensor per micro-batch forward (shared autograd node
            # across layers, like the eager path)
            if "_enc_bias" not in b:
                b["_enc_bias"] = t5.enc_rel_bias(s, s, device)
            return b["_enc_bias"]

        def dec_bias(b, s, device):
            if t5.dec_rel_bias is None:
                return None
            if "_dec_bias" not in b:
                b["_dec_bias"] = t5.dec_rel_bias(s, s, device)
            return b["_dec_bias"]

        def enc_fn(lyr):
            def fn(h, b):
                return t5._run(lyr, h, self._enc_mask(b),
                               position_bias=enc_bias(b, h.size(1), h.device))
            return fn

        for i, layer in enumerate(t5.encoder_layers):
            units.append((i, f"enc_{i}", enc_fn(layer)))
        units.append(
            (n - 1, "enc_final_ln", lambda h, b: t5.encoder_final_layernorm(h))
        )
        units.append(
            (n, "dec_embedding",
             lambda h, b: (h, t5.embedding(b["decoder_input_ids"])))
        )

        def dec_fn(lyr):
            def fn(state, b):
                enc, dec = state
                dec = t5._run(
                    lyr, dec, attention_mask=None, encoder_states=enc,
                    encoder_attention_mask=self._cross_mask(b),
                    position_bias=dec_bias(b, dec.size(1), dec.device),
                )
                return (enc, dec)
            return fn

        for i, layer in enumerate(t5.decoder_layers):
            units.append((n + i, f"dec_{i}", dec_fn(layer)))

        def head(state, b):
            _, dec = state
            h = t5.decoder_final_layernorm(dec)
            logits = t5.lm_head(h, t5.embedding.word_embeddings.weight)
            if b.get("lm_labels") is not None:
                return self.loss_func(logits, b["lm_labels"], b["loss_mask"])
            return {"prediction_scores": logits}

        units.append((-1, "head", head))
        return units

    def pipeline_stage_modules(self):
        t5 = self.t5_model
        n = t5.hidden_layers
        m = {0: [t5.embedding]}
        for i, layer in enumerate(t5.encoder_layers):
            m.setdefault(i, []).append(layer)
            if t5.enc_rel_bias is not None:  # shared bias: every enc stage owns it
                m[i].append(t5.enc_rel_bias)
        m.setdefault(n - 1, []).append(t5.encoder_final_layernorm)
        # the SHARED embedding is owned by the decoder-embedding stage and the
        # head stage too (tied logits); the engine keeps a replica per owning
        # stage and all-reduces its grad over the tied group
        m.setdefault(n, []).append(t5.embedding)
        for i, layer in enumerate(t5.decoder_layers):
            m.setdefault(n + i, []).append(layer)
            if t5.dec_rel_bias is not None:
                m[n + i].append(t5.dec_rel_bias)
        m.setdefault(-1, []).extend(
            [t5.decoder_final_layernorm, t5.lm_head, t5.embedding]
        )
        return m

    def pipeline_boundary_shapes(self, batch, first_idx):
        b, s_enc = batch["encoder_input_ids"].shape
        s_dec = batch["decoder_input_ids"].shape[1]
        h = self.t5_model.hidden_size
        n = self.t5_model.hidden_layers
        if first_idx == -1 or first_idx > n:
            return [(b, s_enc, h), (b, s_dec, h)]
        return [(b, s_enc, h)]
