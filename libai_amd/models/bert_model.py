"""BERT model family (reference: libai/models/bert_model.py:185-556).

attention_mask follows the reference convention at the model boundary:
1 = token is VISIBLE, 0 = padding.  Internally it becomes the [b, sq, sk]
"1 = masked" additive mask the fused softmax consumes (the reference's
BertExtendedAttnMask, bert_model.py:40-60).
"""

import torch
from torch import nn
from torch.utils.checkpoint import checkpoint as act_checkpoint

from ..config import configurable
from ..layers import (
    AttnMaskType,
    Embedding,
    LayerNorm,
    Linear1D,
    LMLogits,
    ParallelCrossEntropyLoss,
    TransformerLayer,
    VocabEmbedding,
    build_activation,
)
from ..utils import distributed as du
from .utils.weight_init import init_method_normal, scaled_init_method_normal

__all__ = ["BertModel", "BertForPreTraining", "BertLoss"]


class BertEmbeddings(nn.Module):
    def __init__(self, vocab_size, hidden_size, max_position_embeddings,
                 num_tokentypes, embedding_dropout_prob, init_method, *, layer_idx=0):
        super().__init__()
        self.vocab_embeddings = VocabEmbedding(vocab_size, hidden_size,
                                               init_method=init_method,
                                               layer_idx=layer_idx)
        self.position_embeddings = Embedding(max_position_embeddings, hidden_size,
                                             init_method=init_method,
                                             layer_idx=layer_idx)
        self.num_tokentypes = num_tokentypes
        if num_tokentypes > 0:
            self.tokentype_embeddings = Embedding(num_tokentypes, hidden_size,
                                                  init_method=init_method,
                                                  layer_idx=layer_idx)
        else:
            self.tokentype_embeddings = None
        self.embedding_dropout = nn.Dropout(embedding_dropout_prob)
        self.register_buffer(
            "position_ids", torch.arange(max_position_embeddings).unsqueeze(0),
            persistent=False,
        )

    def forward(self, input_ids, tokentype_ids=None):
        seq_len = input_ids.size(1)
        emb = self.vocab_embeddings(input_ids)
        emb = emb + self.position_embeddings(self.position_ids[:, :seq_len])
        if self.tokentype_embeddings is not None:
            if tokentype_ids is None:
                tokentype_ids = torch.zeros_like(input_ids)
            emb = emb + self.tokentype_embeddings(tokentype_ids)
        return self.embedding_dropout(emb)


def extended_attn_mask(attention_mask):
    """[b, s] (1=visible) -> [b, sq, sk] uint8 (1=MASKED) via the outer
    product of visibilities (reference: bert_model.py:40-60).

    When the visibility is pure right-padding (a contiguous prefix of 1s)
    the per-sequence valid length is attached as ``mask._kv_len`` — the
    flash-attention kernel consumes that instead of the O(s^2) additive
    mask, so padded BERT/RoBERTa batches stay on the fused path."""
    if attention_mask is None:
        return None
    if attention_mask.dim() == 3:  # pre-built [b, sq, sk] (1=MASKED)
        return attention_mask.to(torch.uint8)
    m = attention_mask.to(torch.uint8)
    visible = m.unsqueeze(1) * m.unsqueeze(2)  # [b, s, s]
    mask = (1 - visible).to(torch.uint8)
    if m.dim() == 2 and (m[:, :-1] >= m[:, 1:]).all():
        mask._kv_len = m.sum(dim=-1, dtype=torch.int32)
    return mask


class BertPooler(nn.Module):
    """Tanh projection of the [CLS] position (reference: bert_model.py)."""

    def __init__(self, hidden_size, init_method, *, layer_idx=-1):
        super().__init__()
        self.dense = Linear1D(hidden_size, hidden_size, parallel="col",
                              init_method=init_method, layer_idx=layer_idx)
        self.activation_func = build_activation("tanh")

    def forward(self, hidden_states):
        from ..parallel.comm import gather_from_tensor_parallel_region

        pooled = self.dense(hidden_states[:, 0])
        pooled = self.activation_func(pooled)
        return gather_from_tensor_parallel_region(pooled)


class BertLMPredictionHead(nn.Module):
    """Transform (dense+gelu+LN) before the tied-embedding logits."""

    def __init__(self, hidden_size, init_method, layernorm_eps, *, layer_idx=-1):
        super().__init__()
        self.dense = Linear1D(hidden_size, hidden_size, parallel="data",
                              init_method=init_method, skip_bias_add=False,
                              layer_idx=layer_idx)
        self.activation_func = build_activation("gelu")
        self.layernorm = LayerNorm(hidden_size, eps=layernorm_eps, layer_idx=layer_idx)

    def forward(self, hidden_states):
        h = self.dense(hidden_states)
        h = self.activation_func(h)
        return self.layernorm(h)


class BertLoss(nn.Module):
    def __init__(self, add_binary_head):
        super().__init__()
        self.add_binary_head = add_binary_head
        self.lm_loss = ParallelCrossEntropyLoss()

    def forward(self, lm_output, lm_labels, loss_mask, binary_logits, ns_labels):
        lm_loss = self.lm_loss(lm_output, lm_labels.clamp(min=0))
        loss_mask = loss_mask.float().view(-1)
        denom = loss_mask.sum().clamp(min=1.0)
        masked_lm_loss = (lm_loss.view(-1) * loss_mask).sum() / denom
        ret = {"lm_loss": masked_lm_loss}
        if self.add_binary_head and binary_logits is not None and ns_labels is not None:
            sop = torch.nn.functional.cross_entropy(binary_logits.float(), ns_labels)
            ret["sop_loss"] = sop
        return ret


class BertModel(nn.Module):
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
        num_tokentypes=2,
        add_pooling_layer=True,
        initializer_range=0.02,
        layernorm_eps=1e-5,
        bias_gelu_fusion=True,
        bias_dropout_fusion=True,
        scale_mask_softmax_fusion=True,
        apply_query_key_layer_scaling=False,
        apply_residual_post_layernorm=False,
        amp_enabled=False,
    ):
        super().__init__()
        init_method = init_method_normal(initializer_range)
        scaled_init = scaled_init_method_normal(initializer_range, hidden_layers)

        self.embeddings = BertEmbeddings(
            vocab_size, hidden_size, max_position_embeddings, num_tokentypes,
            hidden_dropout_prob, init_method, layer_idx=0,
        )
        self.layers = nn.ModuleList(
            [
                TransformerLayer(
                    hidden_size, intermediate_size, num_attention_heads,
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
                    attn_mask_type=AttnMaskType.padding,
                    layer_idx=i,
                )
                for i in range(hidden_layers)
            ]
        )
        self.final_layernorm = LayerNorm(hidden_size, eps=layernorm_eps, layer_idx=-1)
        self.pooler = (
            BertPooler(hidden_size, init_method, layer_idx=-1)
            if add_pooling_layer
            else None
        )
        self.checkpoint_activations = False
        self.hidden_layers = hidden_layers

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
            "num_tokentypes": cfg.get("num_tokentypes", 2),
            "add_pooling_layer": cfg.get("add_pooling_layer", True),
            "initializer_range": cfg.get("initializer_range", 0.02),
            "layernorm_eps": cfg.get("layernorm_eps", 1e-5),
            "bias_gelu_fusion": cfg.get("bias_gelu_fusion", True),
            "bias_dropout_fusion": cfg.get("bias_dropout_fusion", True),
            "scale_mask_softmax_fusion": cfg.get("scale_mask_softmax_fusion", True),
            "apply_query_key_layer_scaling": cfg.get("apply_query_key_layer_scaling", False),
            "apply_residual_post_layernorm": cfg.get("apply_residual_post_layernorm", False),
            "amp_enabled": cfg.get("amp_enabled", False),
        }

    def _run_layer(self, layer, h, mask):
        if self.checkpoint_activations and self.training:
            return act_checkpoint(layer, h, mask, use_reentrant=False)
        return layer(h, attention_mask=mask)

    def forward(self, input_ids, attention_mask=None, tokentype_ids=None):
        mask = extended_attn_mask(attention_mask)
        h = self.embeddings(input_ids, tokentype_ids)
        for layer in self.layers:
            h = self._run_layer(layer, h, mask)
        h = self.final_layernorm(h)
        pooled = self.pooler(h) if self.pooler is not None else None
        return h, pooled

    def set_activation_checkpoint(self, enabled=True):
        self.checkpoint_activations = enabled

    @property
    def word_embeddings_weight(self):
        return self.embeddings.vocab_embeddings.weight


class BertForPreTraining(nn.Module):
    @configurable
    def __init__(self, cfg=None, add_binary_head=True, **kwargs):
        super().__init__()
        if cfg is not None:
            add_binary_head = cfg.get("add_binary_head", True)
            self.bert = BertModel(cfg)
            hidden = cfg.hidden_size
            vocab = cfg.vocab_size
            init_method = init_method_normal(cfg.get("initializer_range", 0.02))
            eps = cfg.get("layernorm_eps", 1e-5)
        else:
            self.bert = BertModel(**kwargs)
            hidden = kwargs["hidden_size"]
            vocab = kwargs["vocab_size"]
            init_method = init_method_normal(kwargs.get("initializer_range", 0.02))
            eps = kwargs.get("layernorm_eps", 1e-5)
        self.cls_head = BertLMPredictionHead(hidden, init_method, eps, layer_idx=-1)
        self.lm_logits = LMLogits(vocab, bias=True, layer_idx=-1)
        self.add_binary_head = add_binary_head
        if add_binary_head:
            self.seq_relationship = Linear1D(hidden, 2, parallel="data",
                                             init_method=init_method, layer_idx=-1)
        self.loss_func = BertLoss(add_binary_head)

    @classmethod
    def from_config(cls, cfg):
        return {"cfg": cfg}

    def forward(self, input_ids, attention_mask=None, tokentype_ids=None,
                ns_labels=None, lm_labels=None, loss_mask=None):
        seq_out, pooled = self.bert(input_ids, attention_mask, tokentype_ids)
        h = self.cls_head(seq_out)
        logits = self.lm_logits(h, self.bert.word_embeddings_weight)
        binary = None
        if self.add_binary_head and pooled is not None:
            binary = self.seq_relationship(pooled)
        if lm_labels is not None and loss_mask is not None:
            return self.loss_func(logits, lm_labels, loss_mask, binary, ns_labels)
        return {"prediction_scores": logits, "seq_relationship_scores": binary}

    def set_activation_checkpoint(self, enabled=True):
        self.bert.set_activation_checkpoint(enabled)

    # -- pipeline protocol --------------------------------------------------

    def pipeline_stage_batch_keys(self, is_first, is_last):
        # every stage rebuilds the padding mask; input_ids ride along as the
        # [b, s] shape witness for the activation boundary
        keys = {"input_ids", "attention_mask"}
        if is_first:
            keys.add("tokentype_ids")
        if is_last:
            keys.update({"lm_labels", "loss_mask", "ns_labels"})
        return keys

    def pipeline_units(self):
        units = [
            (
                0,
                "embeddings",
                lambda h, b: self.bert.embeddings(
                    b["input_ids"], b.get("tokentype_ids")
                ),
            )
        ]
        for i, layer in enumerate(self.bert.layers):
            units.append(
                (
                    i,
                    f"layer_{i}",
                    (
                        lambda lyr: lambda h, b: self.bert._run_layer(
                            lyr, h, extended_attn_mask(b.get("attention_mask"))
                        )
                    )(layer),
                )
            )

        def head(h, b):
            h = self.bert.final_layernorm(h)
            pooled = self.bert.pooler(h) if self.bert.pooler is not None else None
            hh = self.cls_head(h)
            logits = self.lm_logits(hh, self.bert.word_embeddings_weight)
            binary = (
                self.seq_relationship(pooled)
                if self.add_binary_head and pooled is not None
                else None
            )
            if b.get("lm_labels") is not None:
                return self.loss_func(
                    logits, b["lm_labels"], b["loss_mask"], binary, b.get("ns_labels")
                )
            return {"prediction_scores": logits}

        units.append((-1, "head", head))
        return units

    def pipeline_stage_modules(self):
        m = {0: [self.bert.embeddings]}
        for i, layer in enumerate(self.bert.layers):
            m.setdefault(i, []).append(layer)
        last = [self.bert.final_layernorm, self.cls_head, self.lm_logits,
                self.loss_func, self.bert.embeddings.vocab_embeddings]
        if self.bert.pooler is not None:
            last.append(self.bert.pooler)
        if self.add_binary_head:
            last.append(self.seq_relationship)
        m.setdefault(-1, []).extend(last)
        return m
