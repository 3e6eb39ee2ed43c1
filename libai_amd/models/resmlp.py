"""ResMLP (reference: libai/models/resmlp.py): patch embedding + affine-norm
residual blocks mixing across patches (transposed linear) and channels (MLP).
"""

import torch
from torch import nn

from ..config import configurable
from ..layers import DropPath, Linear1D, PatchEmbedding, build_activation
from .utils.weight_init import init_method_normal

__all__ = ["ResMLP"]


class Affine(nn.Module):
    def __init__(self, dim):
        super().__init__()
        self.alpha = nn.Parameter(torch.ones(dim))
        self.beta = nn.Parameter(torch.zeros(dim))

    def forward(self, x):
        return self.alpha * x + self.beta


class ResMLPBlock(nn.Module):
    def __init__(self, dim, num_patches, layerscale_init, drop_path=0.0,
                 *, layer_idx=0):
        super().__init__()
        self.layer_idx = layer_idx
        self.norm1 = Affine(dim)
        self.attn = nn.Linear(num_patches, num_patches)  # cross-patch mixing
        self.gamma_1 = nn.Parameter(layerscale_init * torch.ones(dim))
        self.drop_path = DropPath(drop_path)
        self.norm2 = Affine(dim)
        self.mlp = nn.Sequential(
            Linear1D(dim, 4 * dim, parallel="col",
                     init_method=init_method_normal(0.02), layer_idx=layer_idx),
            build_activation("gelu"),
            Linear1D(4 * dim, dim, parallel="row",
                     init_method=init_method_normal(0.02), layer_idx=layer_idx),
        )
        self.gamma_2 = nn.Parameter(layerscale_init * torch.ones(dim))

    def forward(self, x):
        y = self.attn(self.norm1(x).transpose(1, 2)).transpose(1, 2)
        x = x + self.drop_path(self.gamma_1 * y)
        x = x + self.drop_path(self.gamma_2 * self.mlp(self.norm2(x)))
        return x


class ResMLP(nn.Module):
    @configurable
    def __init__(self, img_size=224, patch_size=16, in_chans=3, embed_dim=768,
                 depth=12, drop_rate=0.0, drop_path_rate=0.0, init_scale=1e-4,
                 num_classes=1000, loss_func=None):
        super().__init__()
        self.patch_embed = PatchEmbedding(img_size=img_size, patch_size=patch_size,
                                          in_chans=in_chans, embed_dim=embed_dim,
                                          layer_idx=0)
        num_patches = self.patch_embed.num_patches
        self.blocks = nn.ModuleList(
            [
                ResMLPBlock(embed_dim, num_patches, init_scale,
                            drop_path=drop_path_rate, layer_idx=i)
                for i in range(depth)
            ]
        )
        self.norm = Affine(embed_dim)
        self.head = Linear1D(embed_dim, num_classes, parallel="data",
                             init_method=init_method_normal(0.02), layer_idx=-1)
        self.loss_func = nn.CrossEntropyLoss() if loss_func is None else loss_func

    @classmethod
    def from_config(cls, cfg):
        return {
            "img_size": cfg.get("img_size", 224),
            "patch_size": cfg.get("patch_size", 16),
            "in_chans": cfg.get("in_chans", 3),
            "embed_dim": cfg.get("embed_dim", 768),
            "depth": cfg.get("depth", 12),
            "drop_rate": cfg.get("drop_rate", 0.0),
            "drop_path_rate": cfg.get("drop_path_rate", 0.0),
            "init_scale": cfg.get("init_scale", 1e-4),
            "num_classes": cfg.get("num_classes", 1000),
            "loss_func": cfg.get("loss_func", None),
        }

    def forward(self, images, labels=None):
        x = self.patch_embed(images)
        for block in self.blocks:
            x = block(x)
        x = self.norm(x).mean(dim=1)
        logits = self.head(x)
        if labels is not None and self.training:
            return {"losses": self.loss_func(logits.float(), labels)}
        return {"prediction_scores": logits}
