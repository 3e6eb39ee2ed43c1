"""RoBERTa model (reference: libai/models/roberta_model.py).

Structurally BERT without the NSP head, with RoBERTa's padding-aware
position ids (positions start at padding_idx+1 and padding tokens keep
padding_idx) and an MLM head (dense + gelu + LN + tied logits).
"""

import torch
from torch import nn

from ..config import configurable
from ..layers import LMLogits, ParallelCrossEntropyLoss
from .bert_model import BertLMPredictionHead, BertModel
from .utils.weight_init import init_method_normal

__all__ = ["RobertaModel", "RobertaForPreTraining", "RobertaForCausalLM"]


def create_position_ids_from_input_ids(input_ids, padding_idx=1):
    mask = input_ids.ne(padding_idx).long()
    incremental = torch.cumsum(mask, dim=1) * mask
    return incremental + padding_idx


class RobertaModel(BertModel):
    """BERT trunk with RoBERTa position-id semantics."""

    @configurable
    def __init__(self, pad_token_id=1, **kwargs):
        super(RobertaModel, self).__init__(**kwargs)
        self.pad_token_id = pad_token_id

    @classmethod
    def from_config(cls, cfg):
        base = BertModel.from_config(cfg)
        base["pad_token_id"] = cfg.get("pad_token_id", 1)
        return base

    def forward(self, input_ids, attention_mask=None, tokentype_ids=None):
        from .bert_model import extended_attn_mask

        mask = extended_attn_mask(attention_mask)
        pos_ids = create_position_ids_from_input_ids(input_ids, self.pad_token_id)
        emb = self.embeddings.vocab_embeddings(input_ids)
        emb = emb + self.embeddings.position_embeddings(
            pos_ids.clamp(max=self.embeddings.position_embeddings.num_embeddings - 1)
        )
        if self.embeddings.tokentype_embeddings is not None:
            if tokentype_ids is None:
                tokentype_ids = torch.zeros_like(input_ids)
            emb = emb + self.embeddings.tokentype_embeddings(tokentype_ids)
        h = self.embeddings.embedding_dropout(emb)
        for layer in self.layers:
            h = self._run_layer(layer, h, mask)
        h = self.final_layernorm(h)
        pooled = self.pooler(h) if self.pooler is not None else None
        return h, pooled


class RobertaLoss(nn.Module):
    def __init__(self):
        super().__init__()
        self.lm_loss = ParallelCrossEntropyLoss()

    def forward(self, logits, lm_labels, loss_mask):
        loss = self.lm_loss(logits, lm_labels.clamp(min=0)).view(-1)
        lm = loss * loss_mask.float().view(-1)
        return {"lm_loss": lm.sum() / loss_mask.float().sum().clamp(min=1.0)}


class RobertaForPreTraining(nn.Module):
    @configurable
    def __init__(self, cfg=None, **kwargs):
        super().__init__()
        if cfg is not None:
            self.roberta = RobertaModel(cfg)
            hidden = cfg.hidden_size
            vocab = cfg.vocab_size
            init_method = init_method_normal(cfg.get("initializer_range", 0.02))
            eps = cfg.get("layernorm_eps", 1e-5)
        else:
            self.roberta = RobertaModel(**kwargs)
            hidden = kwargs["hidden_size"]
            vocab = kwargs["vocab_size"]
            init_method = init_method_normal(kwargs.get("initializer_range", 0.02))
            eps = kwargs.get("layernorm_eps", 1e-5)
        self.lm_head = BertLMPredictionHead(hidden, init_method, eps, layer_idx=-1)
        self.lm_logits = LMLogits(vocab, bias=True, layer_idx=-1)
        self.loss_func = RobertaLoss()

    @classmethod
    def from_config(cls, cfg):
        return {"cfg": cfg}

    def forward(self, input_ids, attention_mask=None, tokentype_ids=None,
                lm_labels=None, loss_mask=None):
        seq_out, _ = self.roberta(input_ids, attention_mask, tokentype_ids)
        h = self.lm_head(seq_out)
        logits = self.lm_logits(h, self.roberta.word_embeddings_weight)
        if lm_labels is not None and loss_mask is not None:
            return self.loss_func(logits, lm_labels, loss_mask)
        return {"prediction_scores": logits}

    def set_activation_checkpoint(self, enabled=True):
        self.roberta.set_activation_checkpoint(enabled)

    # -- pipeline-parallel protocol (reference assigns RoBERTa stage ids in
    # roberta_model.py; single-tensor boundary like BERT) -------------------

    def _embed(self, b):
        rb = self.roberta
        input_ids = b["input_ids"]
        pos_ids = create_position_ids_from_input_ids(input_ids, rb.pad_token_id)
        emb = rb.embeddings.vocab_embeddings(input_ids)
        emb = emb + rb.embeddings.position_embeddings(
            pos_ids.clamp(max=rb.embeddings.position_embeddings.num_embeddings - 1)
        )
        if rb.embeddings.tokentype_embeddings is not None:
            tt = b.get("tokentype_ids")
            if tt is None:
                tt = torch.zeros_like(input_ids)
            emb = emb + rb.embeddings.tokentype_embeddings(tt)
        return rb.embeddings.embedding_dropout(emb)

    def pipeline_stage_batch_keys(self, is_first, is_last):
        keys = {"input_ids", "attention_mask"}
        if is_first:
            keys.add("tokentype_ids")
        if is_last:
            keys.update({"lm_labels", "loss_mask"})
        return keys

    def pipeline_units(self):
        from .bert_model import extended_attn_mask

        rb = self.roberta
        units = [(0, "embeddings", lambda h, b: self._embed(b))]
        for i, layer in enumerate(rb.layers):
            units.append(
                (i, f"layer_{i}",
                 (lambda lyr: lambda h, b: rb._run_layer(
                     lyr, h, extended_attn_mask(b.get("attention_mask"))
                 ))(layer))
            )

        def head(h, b):
            h = rb.final_layernorm(h)
            hh = self.lm_head(h)
            logits = self.lm_logits(hh, rb.word_embeddings_weight)
            if b.get("lm_labels") is not None and b.get("loss_mask") is not None:
                return self.loss_func(logits, b["lm_labels"], b["loss_mask"])
            return {"prediction_scores": logits}

        units.append((-1, "head", head))
        return units

    def pipeline_stage_modules(self):
        rb = self.roberta
        m = {0: [rb.embeddings]}
        for i, layer in enumerate(rb.layers):
            m.setdefault(i, []).append(layer)
        last = [rb.final_layernorm, self.lm_head, self.lm_logits, self.loss_func,
                rb.embeddings.vocab_embeddings]
        if rb.pooler is not None:  # unused by the pretraining loss, but owned
            last.append(rb.pooler)  # by the last stage so other stages free it
        m.setdefault(-1, []).extend(last)
        return m


class RobertaForCausalLM(RobertaForPreTraining):
    """RoBERTa as a left-to-right LM (reference: roberta_model.py's CLM
    variant): the padding mask is combined with a causal mask and the loss
    is next-token CE over the shifted sequence."""

    def forward(self, input_ids, attention_mask=None, tokentype_ids=None,
                labels=None, **kwargs):
        b, s = input_ids.shape
        causal = torch.tril(
            torch.ones(s, s, dtype=torch.bool, device=input_ids.device)
        )
        if attention_mask is not None:
            vis = attention_mask.to(torch.bool)[:, None, :] & causal[None]
        else:
            vis = causal[None].expand(b, s, s)
        # RobertaModel's extended_attn_mask passes a prebuilt [b, s, s]
        # (1 = MASKED) through unchanged
        masked = (~vis).to(torch.uint8)
        seq_out, _ = self.roberta(input_ids, masked, tokentype_ids)
        h = self.lm_head(seq_out)
        logits = self.lm_logits(h, self.roberta.word_embeddings_weight)
        if labels is not None:
            shift_logits = logits[:, :-1].contiguous()
            shift_labels = labels[:, 1:].contiguous()
            mask = torch.ones_like(shift_labels)
            if attention_mask is not None:
                mask = attention_mask[:, 1:].to(mask.dtype)
            out = self.loss_func(shift_logits, shift_labels, mask)
            return {"lm_loss": out["lm_loss"]}
        return {"prediction_scores": logits}
