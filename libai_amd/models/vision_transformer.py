"""Vision Transformer (reference: libai/models/vision_transformer.py:49-267)."""

import torch
from torch import nn
from torch.utils.checkpoint import checkpoint as act_checkpoint

from ..config import configurable
from ..layers import LayerNorm, Linear1D, PatchEmbedding, TransformerLayer
from .utils.weight_init import init_method_normal

__all__ = ["VisionTransformer", "ViTEmbedding"]


class ViTEmbedding(nn.Module):
    """Patchify + cls token + learned positions (stage-0 pipeline unit)."""

    def __init__(self, img_size, patch_size, in_chans, embed_dim, drop_rate,
                 *, layer_idx=0):
        super().__init__()
        self.patch_embed = PatchEmbedding(
            img_size=img_size, patch_size=patch_size, in_chans=in_chans,
            embed_dim=embed_dim, layer_idx=layer_idx,
        )
        self.num_patches = self.patch_embed.num_patches
        self.cls_token = nn.Parameter(torch.zeros(1, 1, embed_dim))
        self.pos_embed = nn.Parameter(torch.zeros(1, self.num_patches + 1, embed_dim))
        nn.init.trunc_normal_(self.pos_embed, std=0.02)
        nn.init.trunc_normal_(self.cls_token, std=0.02)
        self.pos_drop = nn.Dropout(drop_rate)

    def forward(self, images):
        x = self.patch_embed(images)
        cls = self.cls_token.expand(x.shape[0], -1, -1).to(x.dtype)
        x = torch.cat((cls, x), dim=1)
        return self.pos_drop(x + self.pos_embed.to(x.dtype))


class VisionTransformer(nn.Module):
    @configurable
    def __init__(
        self,
        img_size=224,
        patch_size=16,
        in_chans=3,
        embed_dim=192,
        depth=12,
        num_heads=3,
        mlp_ratio=4.0,
        drop_rate=0.0,
        attn_drop_rate=0.0,
        drop_path_rate=0.0,
        num_classes=1000,
        loss_func=None,
    ):
        super().__init__()
        self.img_size = img_size
        self.num_classes = num_classes
        self.depth = depth
        self.embed_dim = embed_dim
        init_method = init_method_normal(0.02)

        self.embedding = ViTEmbedding(img_size, patch_size, in_chans, embed_dim,
                                      drop_rate)
        self.patch_embed_seq_len = self.embedding.num_patches + 1  # + CLS

        ffn_size = int(embed_dim * mlp_ratio)
        self.blocks = nn.ModuleList(
            [
                TransformerLayer(
                    embed_dim, ffn_size, num_heads,
                    attention_dropout_prob=attn_drop_rate,
                    output_dropout_prob=drop_rate,
                    init_method=init_method,
                    attn_mask_type="padding",
                    layer_idx=i,
                )
                for i in range(depth)
            ]
        )
        self.norm = LayerNorm(embed_dim, layer_idx=-1)
        self.head = Linear1D(embed_dim, num_classes, parallel="data",
                             init_method=init_method, layer_idx=-1)
        self.loss_func = nn.CrossEntropyLoss() if loss_func is None else loss_func
        self.checkpoint_activations = False

    @classmethod
    def from_config(cls, cfg):
        return {
            "img_size": cfg.get("img_size", 224),
            "patch_size": cfg.get("patch_size", 16),
            "in_chans": cfg.get("in_chans", 3),
            "embed_dim": cfg.get("embed_dim", 192),
            "depth": cfg.get("depth", 12),
            "num_heads": cfg.get("num_heads", 3),
            "mlp_ratio": cfg.get("mlp_ratio", 4.0),
            "drop_rate": cfg.get("drop_rate", 0.0),
            "attn_drop_rate": cfg.get("attn_drop_rate", 0.0),
            "drop_path_rate": cfg.get("drop_path_rate", 0.0),
            "num_classes": cfg.get("num_classes", 1000),
            "loss_func": cfg.get("loss_func", None),
        }

    def forward_features(self, x):
        x = self.embedding(x)
        for block in self.blocks:
            if self.checkpoint_activations and self.training:
                x = act_checkpoint(block, x, use_reentrant=False)
            else:
                x = block(x)
        return self.norm(x)

    def forward_head(self, x):
        return self.head(x[:, 0])

    def forward(self, images, labels=None):
        x = self.forward_features(images)
        logits = self.forward_head(x)
        if labels is not None and self.training:
            return {"losses": self.loss_func(logits.float(), labels)}
        return {"prediction_scores": logits}

    def set_activation_checkpoint(self, enabled=True):
        self.checkpoint_activations = enabled

    def pipeline_stage_batch_keys(self, is_first, is_last):
        # labels ([b] ints) ride along everywhere as the batch-size witness
        # for boundary shapes; the [b, 3, H, W] images only go to stage 0
        keys = {"labels"}
        if is_first:
            keys.add("images")
        return keys

    def pipeline_boundary_shapes(self, batch, first_idx):
        ref = batch.get("images", batch.get("labels"))
        b = ref.shape[0]
        return [(b, self.patch_embed_seq_len, self.embed_dim)]

    # -- pipeline protocol --------------------------------------------------

    def pipeline_units(self):
        units = [(0, "embedding", lambda h, b: self.embedding(b["images"]))]
        for i, block in enumerate(self.blocks):
            units.append((i, f"block_{i}", (lambda m: lambda h, b: m(h))(block)))

        def head(h, b):
            h = self.norm(h)
            logits = self.head(h[:, 0])
            if b.get("labels") is not None and self.training:
                return {"losses": self.loss_func(logits.float(), b["labels"])}
            return {"prediction_scores": logits}

        units.append((-1, "head", head))
        return units

    def pipeline_stage_modules(self):
        m = {0: [self.embedding]}
        for i, block in enumerate(self.blocks):
            m.setdefault(i, []).append(block)
        m.setdefault(-1, []).extend([self.norm, self.head])
        return m
