"""Model loaders: load LiBai-AMD or HuggingFace checkpoints into a (possibly
TP-sharded) model.

Reference behavior: libai/models/utils/model_loader/base_loader.py:90-140 —
read full tensors, remap keys per architecture, shard per-tensor onto the
current topology.  Sharding metadata comes from the target parameters'
``tensor_parallel`` / ``tp_shard_dim`` attributes, so any architecture built
from libai_amd.layers reshard-loads without per-arch shard tables.
"""

import logging
import os

import torch

from ...utils import distributed as du

__all__ = ["ModelLoaderLiBai", "ModelLoaderHuggerFace", "GPT2LoaderHuggerFace",
           "BertLoaderHuggerFace", "LlamaLoaderHuggerFace"]

logger = logging.getLogger(__name__)


def _shard_and_load(model, full_state, strict=False):
    dutil = du.get_dist_util()
    tp, tpr = dutil.tensor_parallel_size, dutil.tensor_parallel_rank
    params = dict(model.named_parameters())
    local = {}
    for name, cur in model.state_dict().items():
        if name not in full_state:
            continue
        t = full_state[name]
        p = params.get(name)
        if p is not None and getattr(p, "tensor_parallel", False) and tp > 1:
            from ...layers.linear import tp_slice

            t = tp_slice(t, tp, tpr, getattr(p, "tp_shard_dim", 0),
                         getattr(p, "tp_fused_chunks", 1))
        if tuple(t.shape) != tuple(cur.shape):
            logger.warning(
                f"loader: {name} shape {tuple(t.shape)} != model {tuple(cur.shape)}"
            )
            continue
        local[name] = t.to(cur.dtype)
    missing, unexpected = model.load_state_dict(local, strict=False)
    param_missing = [m for m in missing if m in params]
    if param_missing:
        logger.warning(f"loader missing keys: {param_missing[:12]}")
    if strict and param_missing:
        raise KeyError(f"missing keys: {param_missing}")
    return model


class ModelLoaderLiBai:
    """Load a libai_amd checkpoint directory (model.pt with full tensors)."""

    def __init__(self, model, libai_cfg=None, pretrained_model_path="", **kwargs):
        self.model = model
        self.path = pretrained_model_path

    def load(self):
        f = self.path
        if os.path.isdir(f):
            f = os.path.join(f, "model.pt")
        state = torch.load(f, map_location="cpu", weights_only=False)
        return _shard_and_load(self.model, state)


class ModelLoaderHuggerFace:
    """Load HF pytorch_model.bin / model.safetensors with key remapping."""

    base_model_prefix_1 = ""  # HF prefix
    base_model_prefix_2 = ""  # our prefix

    def __init__(self, model, libai_cfg=None, pretrained_model_path="", **kwargs):
        self.model = model
        self.cfg = libai_cfg
        self.path = pretrained_model_path

    def _read_state(self):
        p = self.path
        candidates = []
        if os.path.isdir(p):
            for name in os.listdir(p):
                if name.endswith(".safetensors") or name.endswith(".bin"):
                    candidates.append(os.path.join(p, name))
        else:
            candidates = [p]
        state = {}
        for f in sorted(candidates):
            if f.endswith(".safetensors"):
                from safetensors.torch import load_file

                state.update(load_file(f))
            else:
                state.update(torch.load(f, map_location="cpu", weights_only=False))
        return state

    def _convert_state_dict(self, hf_state):
        """Subclasses remap HF keys to libai_amd keys."""
        return hf_state

    def load(self):
        state = self._convert_state_dict(self._read_state())
        return _shard_and_load(self.model, state)


class GPT2LoaderHuggerFace(ModelLoaderHuggerFace):
    """HF GPT-2 -> libai_amd GPTModel.

    HF stores qkv as Conv1D [in, 3h] ordered (q|k|v) over the WHOLE hidden;
    our fused qkv weight is [3h, in] ordered per-head (q|k|v within each
    head), so the conversion re-interleaves heads.
    """

    def _convert_state_dict(self, hf):
        cfg = self.cfg
        nh = cfg.num_attention_heads if cfg is not None else None
        out = {}
        prefix = "GPT_model."
        strip = lambda k: k[len("transformer."):] if k.startswith("transformer.") else k
        for k, v in hf.items():
            k = strip(k)
            if k == "wte.weight":
                out[prefix + "embeddings.token_embeddings.weight"] = v
            elif k == "wpe.weight":
                out[prefix + "embeddings.position_embeddings.weight"] = v
            elif k == "ln_f.weight":
                out[prefix + "transformer.layernorm_f.weight"] = v
            elif k == "ln_f.bias":
                out[prefix + "transformer.layernorm_f.bias"] = v
            elif k.startswith("h."):
                parts = k.split(".")
                i, rest = parts[1], ".".join(parts[2:])
                base = f"{prefix}transformer.layers.{i}."
                if rest == "ln_1.weight":
                    out[base + "input_layernorm.weight"] = v
                elif rest == "ln_1.bias":
                    out[base + "input_layernorm.bias"] = v
                elif rest == "ln_2.weight":
                    out[base + "post_attention_layernorm.weight"] = v
                elif rest == "ln_2.bias":
                    out[base + "post_attention_layernorm.bias"] = v
                elif rest == "attn.c_attn.weight":
                    w = v.t().contiguous()  # [3h, h]
                    out[base + "self_attention.query_key_value.weight"] = \
                        self._interleave_qkv(w, nh)
                elif rest == "attn.c_attn.bias":
                    out[base + "self_attention.query_key_value.bias"] = \
                        self._interleave_qkv(v, nh)
                elif rest == "attn.c_proj.weight":
                    out[base + "self_attention.dense.weight"] = v.t().contiguous()
                elif rest == "attn.c_proj.bias":
                    out[base + "self_attention.dense.bias"] = v
                elif rest == "mlp.c_fc.weight":
                    out[base + "mlp.dense_h_to_4h.weight"] = v.t().contiguous()
                elif rest == "mlp.c_fc.bias":
                    out[base + "mlp.dense_h_to_4h.bias"] = v
                elif rest == "mlp.c_proj.weight":
                    out[base + "mlp.dense_4h_to_h.weight"] = v.t().contiguous()
                elif rest == "mlp.c_proj.bias":
                    out[base + "mlp.dense_4h_to_h.bias"] = v
        return out

    @staticmethod
    def _interleave_qkv(w, num_heads):
        """[3h, ...] ordered (q_all|k_all|v_all) -> per-head (q|k|v)."""
        if num_heads is None:
            return w
        three_h = w.shape[0]
        h = three_h // 3
        hs = h // num_heads
        q, k, v = w[:h], w[h : 2 * h], w[2 * h :]
        chunks = []
        for head in range(num_heads):
            sl = slice(head * hs, (head + 1) * hs)
            chunks.extend([q[sl], k[sl], v[sl]])
        return torch.cat(chunks, dim=0)


class BertLoaderHuggerFace(ModelLoaderHuggerFace):
    def _convert_state_dict(self, hf):
        cfg = self.cfg
        nh = cfg.num_attention_heads if cfg is not None else None
        out = {}
        for k, v in hf.items():
            k = k[len("bert."):] if k.startswith("bert.") else k
            if k == "embeddings.word_embeddings.weight":
                out["bert.embeddings.vocab_embeddings.weight"] = v
            elif k == "embeddings.position_embeddings.weight":
                out["bert.embeddings.position_embeddings.weight"] = v
            elif k == "embeddings.token_type_embeddings.weight":
                out["bert.embeddings.tokentype_embeddings.weight"] = v
            elif k.startswith("encoder.layer."):
                parts = k.split(".")
                i, rest = parts[2], ".".join(parts[3:])
                base = f"bert.layers.{i}."
                m = {
                    "attention.output.dense.weight": "self_attention.dense.weight",
                    "attention.output.dense.bias": "self_attention.dense.bias",
                    "intermediate.dense.weight": "mlp.dense_h_to_4h.weight",
                    "intermediate.dense.bias": "mlp.dense_h_to_4h.bias",
                    "output.dense.weight": "mlp.dense_4h_to_h.weight",
                    "output.dense.bias": "mlp.dense_4h_to_h.bias",
                    "attention.output.LayerNorm.weight": "post_attention_layernorm.weight",
                    "attention.output.LayerNorm.bias": "post_attention_layernorm.bias",
                }
                if rest in m:
                    out[base + m[rest]] = v
                # HF computes LN after sublayers (post-LN); our blocks are
                # pre-LN, so q/k/v fusion is handled below and exact HF
                # equivalence is approximate for fine-tuning starts.
        # fuse q/k/v per layer
        layers = {}
        for k, v in hf.items():
            k2 = k[len("bert."):] if k.startswith("bert.") else k
            if ".attention.self." in k2:
                parts = k2.split(".")
                i = parts[2]
                which = parts[4]  # query/key/value
                kind = parts[5]  # weight/bias
                layers.setdefault((i, kind), {})[which] = v
        for (i, kind), d in layers.items():
            if len(d) == 3 and nh is not None:
                stacked = torch.cat([d["query"], d["key"], d["value"]], dim=0)
                out[f"bert.layers.{i}.self_attention.query_key_value.{kind}"] = \
                    GPT2LoaderHuggerFace._interleave_qkv(stacked, nh)
        return out


class LlamaLoaderHuggerFace(ModelLoaderHuggerFace):
    def _convert_state_dict(self, hf):
        cfg = self.cfg
        nh = cfg.num_attention_heads if cfg is not None else None
        out = {}
        qkv = {}
        for k, v in hf.items():
            k = k[len("model."):] if k.startswith("model.") else k
            if k == "embed_tokens.weight":
                out["model.embed_tokens.weight"] = v
            elif k == "norm.weight":
                out["model.norm.weight"] = v
            elif k == "lm_head.weight":
                out["lm_head.weight"] = v
            elif k.startswith("layers."):
                parts = k.split(".")
                i, rest = parts[1], ".".join(parts[2:])
                base = f"model.layers.{i}."
                if rest in ("input_layernorm.weight", "post_attention_layernorm.weight"):
                    out[base + rest] = v
                elif rest == "self_attn.o_proj.weight":
                    out[base + "self_attn.o_proj.weight"] = v
                elif rest in ("self_attn.q_proj.weight", "self_attn.k_proj.weight",
                              "self_attn.v_proj.weight"):
                    qkv.setdefault(i, {})[rest.split(".")[1][0]] = v
                elif rest == "mlp.gate_proj.weight":
                    qkv.setdefault(i, {})["g"] = v
                elif rest == "mlp.up_proj.weight":
                    qkv.setdefault(i, {})["u"] = v
                elif rest == "mlp.down_proj.weight":
                    out[base + "mlp.down_proj.weight"] = v
        nkv = None
        if cfg is not None:
            nkv = getattr(cfg, "num_key_value_heads", None)
            if nkv is None and hasattr(cfg, "get"):
                nkv = cfg.get("num_key_value_heads", None)
            nkv = nkv or nh
        for i, d in qkv.items():
            base = f"model.layers.{i}."
            if all(x in d for x in "qkv"):
                if nkv is not None and nkv != nh:
                    # GQA (Llama-70B/Qwen2 class): separate q + fused [k|v]
                    out[base + "self_attn.q_proj.weight"] = d["q"]
                    out[base + "self_attn.kv_proj.weight"] = \
                        torch.cat([d["k"], d["v"]], dim=0)
                else:
                    stacked = torch.cat([d["q"], d["k"], d["v"]], dim=0)
                    out[base + "self_attn.query_key_value.weight"] = \
                        GPT2LoaderHuggerFace._interleave_qkv(stacked, nh)
            if "g" in d and "u" in d:
                out[base + "mlp.gate_up_proj.weight"] = torch.cat([d["g"], d["u"]], 0)
        return out


class ViTLoaderHuggerFace(ModelLoaderHuggerFace):
    """HF ViT (google/vit-*) -> libai_amd VisionTransformer.

    Reference: model_loader/vit_loader.py.  HF stores separate q/k/v
    projections; our attention uses the fused per-head-interleaved qkv.
    """

    def _convert_state_dict(self, hf):
        cfg = self.cfg
        nh = getattr(cfg, "num_heads", None) if cfg is not None else None
        out = {}
        qkv = {}
        for k, v in hf.items():
            k = k[len("vit."):] if k.startswith("vit.") else k
            if k == "embeddings.cls_token":
                out["embedding.cls_token"] = v
            elif k == "embeddings.position_embeddings":
                out["embedding.pos_embed"] = v
            elif k == "embeddings.patch_embeddings.projection.weight":
                out["embedding.patch_embed.proj.weight"] = v
            elif k == "embeddings.patch_embeddings.projection.bias":
                out["embedding.patch_embed.proj.bias"] = v
            elif k == "layernorm.weight":
                out["norm.weight"] = v
            elif k == "layernorm.bias":
                out["norm.bias"] = v
            elif k == "classifier.weight":
                out["head.weight"] = v
            elif k == "classifier.bias":
                out["head.bias"] = v
            elif k.startswith("encoder.layer."):
                parts = k.split(".")
                i, rest = parts[2], ".".join(parts[3:])
                base = f"blocks.{i}."
                m = {
                    "layernorm_before.weight": "input_layernorm.weight",
                    "layernorm_before.bias": "input_layernorm.bias",
                    "layernorm_after.weight": "post_attention_layernorm.weight",
                    "layernorm_after.bias": "post_attention_layernorm.bias",
                    "attention.output.dense.weight": "self_attention.dense.weight",
                    "attention.output.dense.bias": "self_attention.dense.bias",
                    "intermediate.dense.weight": "mlp.dense_h_to_4h.weight",
                    "intermediate.dense.bias": "mlp.dense_h_to_4h.bias",
                    "output.dense.weight": "mlp.dense_4h_to_h.weight",
                    "output.dense.bias": "mlp.dense_4h_to_h.bias",
                }
                if rest in m:
                    out[base + m[rest]] = v
                elif rest.startswith("attention.attention."):
                    which = rest.split(".")[2]  # query/key/value
                    kind = rest.split(".")[3]
                    qkv.setdefault((i, kind), {})[which] = v
        for (i, kind), d in qkv.items():
            if len(d) == 3:
                stacked = torch.cat([d["query"], d["key"], d["value"]], dim=0)
                out[f"blocks.{i}.self_attention.query_key_value.{kind}"] = \
                    GPT2LoaderHuggerFace._interleave_qkv(stacked, nh)
        return out


class RobertaLoaderHuggerFace(BertLoaderHuggerFace):
    """HF RoBERTa -> libai_amd RobertaModel (reference: roberta_loader.py).

    RoBERTa is BERT-shaped with a `roberta.` prefix and no NSP head; reuse
    the BERT mapping after re-prefixing the keys onto the `roberta.` module.
    """

    def _convert_state_dict(self, hf):
        renamed = {}
        for k, v in hf.items():
            if k.startswith("roberta."):
                k = "bert." + k[len("roberta."):]
            renamed[k] = v
        bert_out = super()._convert_state_dict(renamed)
        return {
            ("roberta." + k[len("bert."):]) if k.startswith("bert.") else k: v
            for k, v in bert_out.items()
        }


class SwinLoaderHuggerFace(ModelLoaderHuggerFace):
    """HF Swin (microsoft/swin-*) -> libai_amd SwinTransformer.

    Reference: model_loader/swin_loader.py.  HF stores per-block modules
    under encoder.layers.{i}.blocks.{j}; ours are layers.{i}.0.{j} with the
    patch-merging downsample at layers.{i}.1.  HF's separate q/k/v stack
    into our fused qkv (q|k|v over the whole width; window attention is
    not head-interleaved).
    """

    def _convert_state_dict(self, hf):
        out = {}
        qkv = {}
        blk = {
            "layernorm_before.weight": "norm1.weight",
            "layernorm_before.bias": "norm1.bias",
            "layernorm_after.weight": "norm2.weight",
            "layernorm_after.bias": "norm2.bias",
            "attention.self.relative_position_bias_table":
                "attn.relative_position_bias_table",
            "attention.output.dense.weight": "attn.proj.weight",
            "attention.output.dense.bias": "attn.proj.bias",
            "intermediate.dense.weight": "mlp.0.weight",
            "intermediate.dense.bias": "mlp.0.bias",
            "output.dense.weight": "mlp.3.weight",
            "output.dense.bias": "mlp.3.bias",
        }
        for k, v in hf.items():
            if k.startswith("swin."):
                k = k[len("swin."):]
            if k == "embeddings.patch_embeddings.projection.weight":
                out["patch_embed.weight"] = v
            elif k == "embeddings.patch_embeddings.projection.bias":
                out["patch_embed.bias"] = v
            elif k == "embeddings.norm.weight":
                out["patch_norm.weight"] = v
            elif k == "embeddings.norm.bias":
                out["patch_norm.bias"] = v
            elif k == "layernorm.weight":
                out["norm.weight"] = v
            elif k == "layernorm.bias":
                out["norm.bias"] = v
            elif k == "classifier.weight":
                out["head.weight"] = v
            elif k == "classifier.bias":
                out["head.bias"] = v
            elif k.startswith("encoder.layers."):
                parts = k.split(".")
                i = parts[2]
                if parts[3] == "blocks":
                    j, rest = parts[4], ".".join(parts[5:])
                    base = f"layers.{i}.0.{j}."
                    if rest in blk:
                        out[base + blk[rest]] = v
                    elif rest.startswith("attention.self."):
                        which = rest.split(".")[2]  # query/key/value
                        kind = rest.split(".")[3]
                        qkv.setdefault((i, j, kind), {})[which] = v
                elif parts[3] == "downsample":
                    rest = ".".join(parts[4:])
                    m = {"reduction.weight": "reduction.weight",
                         "norm.weight": "norm.weight", "norm.bias": "norm.bias"}
                    if rest in m:
                        out[f"layers.{i}.1.{m[rest]}"] = v
        for (i, j, kind), d in qkv.items():
            if len(d) == 3:
                out[f"layers.{i}.0.{j}.attn.qkv.{kind}"] = torch.cat(
                    [d["query"], d["key"], d["value"]], dim=0
                )
        return out


class SwinV2LoaderHuggerFace(SwinLoaderHuggerFace):
    """HF SwinV2 (microsoft/swinv2-*) -> libai_amd SwinTransformerV2.

    Reference: model_loader/swinv2_loader.py.  On top of the Swin mapping,
    V2 adds the per-head logit scale and the continuous-position-bias MLP
    (HF: continuous_position_bias_mlp / logit_scale).
    """

    def _convert_state_dict(self, hf):
        out = super()._convert_state_dict(hf)
        for k, v in hf.items():
            if k.startswith("swin."):
                k = k[len("swin."):]
            elif k.startswith("swinv2."):
                k = k[len("swinv2."):]
            if not k.startswith("encoder.layers."):
                continue
            parts = k.split(".")
            i = parts[2]
            if parts[3] != "blocks":
                continue
            j, rest = parts[4], ".".join(parts[5:])
            base = f"layers.{i}.0.{j}.attn."
            if rest == "attention.self.logit_scale":
                out[base + "logit_scale"] = v
            elif rest.startswith(
                "attention.self.continuous_position_bias_mlp."
            ):
                idx_rest = rest[len("attention.self.continuous_position_bias_mlp."):]
                out[base + "cpb_mlp." + idx_rest] = v
        return out
