"""CLIP: dual-tower image/text contrastive pretraining + inference.

Reference capability: projects/CLIP (CLIP inference/finetune built on the
library).  Image tower = ViT (CLS feature), text tower = causal transformer
pooled at the end-of-text token, learned temperature, symmetric InfoNCE.
"""

import math

import torch
import torch.nn.functional as F
from torch import nn

from ..config import configurable
from ..layers import AttnMaskType, Embedding, LayerNorm, TransformerLayer, VocabEmbedding
from .utils.weight_init import init_method_normal
from .vision_transformer import VisionTransformer

__all__ = ["CLIPModel"]


class _TextTower(nn.Module):
    def __init__(self, vocab_size, width, layers, heads, context_length,
                 init_method):
        super().__init__()
        self.token_embedding = VocabEmbedding(vocab_size, width,
                                              init_method=init_method)
        self.position_embedding = Embedding(context_length, width,
                                            init_method=init_method)
        self.blocks = nn.ModuleList([
            TransformerLayer(width, 4 * width, heads,
                             attn_mask_type=AttnMaskType.causal,
                             init_method=init_method, layer_idx=i)
            for i in range(layers)
        ])
        self.ln_final = LayerNorm(width)
        self.register_buffer("pos_ids",
                             torch.arange(context_length).unsqueeze(0),
                             persistent=False)

    def forward(self, text_ids):
        s = text_ids.shape[1]
        h = self.token_embedding(text_ids) + self.position_embedding(
            self.pos_ids[:, :s])
        for blk in self.blocks:
            h = blk(h)
        h = self.ln_final(h)
        # pool at the EOT token = the argmax token id (CLIP convention:
        # the eot id is the largest id in the vocab)
        eot = text_ids.argmax(dim=-1)
        return h[torch.arange(h.shape[0], device=h.device), eot]


class CLIPModel(nn.Module):
    @configurable
    def __init__(
        self,
        embed_dim=512,
        # vision
        img_size=224,
        patch_size=16,
        vision_width=768,
        vision_layers=12,
        vision_heads=12,
        # text
        vocab_size=49408,
        context_length=77,
        text_width=512,
        text_layers=12,
        text_heads=8,
        initializer_range=0.02,
    ):
        super().__init__()
        init_method = init_method_normal(initializer_range)
        self.visual = VisionTransformer(
            img_size=img_size, patch_size=patch_size, embed_dim=vision_width,
            depth=vision_layers, num_heads=vision_heads, num_classes=8,
        )
        self.text = _TextTower(vocab_size, text_width, text_layers, text_heads,
                               context_length, init_method)
        self.visual_projection = nn.Linear(vision_width, embed_dim, bias=False)
        self.text_projection = nn.Linear(text_width, embed_dim, bias=False)
        self.logit_scale = nn.Parameter(
            torch.tensor(math.log(1 / 0.07), dtype=torch.float32))

    @classmethod
    def from_config(cls, cfg):
        return {k: cfg.get(k) for k in (
            "embed_dim", "img_size", "patch_size", "vision_width",
            "vision_layers", "vision_heads", "vocab_size", "context_length",
            "text_width", "text_layers", "text_heads", "initializer_range",
        ) if cfg.get(k) is not None}

    def encode_image(self, images):
        feats = self.visual.forward_features(images)[:, 0]
        return F.normalize(self.visual_projection(feats), dim=-1)

    def encode_text(self, text_ids):
        return F.normalize(self.text_projection(self.text(text_ids)), dim=-1)

    def forward(self, images=None, text_ids=None, labels=None):
        if images is not None and text_ids is None:
            return {"image_embeds": self.encode_image(images)}
        if text_ids is not None and images is None:
            return {"text_embeds": self.encode_text(text_ids)}
        zi = self.encode_image(images)
        zt = self.encode_text(text_ids)
        scale = self.logit_scale.exp().clamp(max=100.0)
        logits = scale * zi.float() @ zt.float().t()  # [b, b]
        if self.training:
            target = torch.arange(zi.shape[0], device=zi.device)
            loss = 0.5 * (F.cross_entropy(logits, target)
                          + F.cross_entropy(logits.t(), target))
            return {"clip_loss": loss}
        return {"logits_per_image": logits, "logits_per_text": logits.t()}
