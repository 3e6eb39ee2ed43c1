"""ConvNeXt image classifier (reference capability: projects/ConvNeXT —
ConvNeXt on the library, DP training like Swin).

Standard ConvNeXt blocks: 7x7 depthwise conv -> channels-last LayerNorm ->
pointwise 4x expand -> GELU -> pointwise project, with LayerScale and
stochastic depth (DropPath).  Downsample stems between stages.  DP-parallel
(conv trunks don't TP-shard usefully); the engine's DDP/ZeRO handles
scaling, and the classification head/loss match the ViT recipe so the same
loaders/evaluators apply.
"""

import torch
from torch import nn

from ..config import configurable
from ..layers import DropPath

__all__ = ["ConvNeXt"]


class _Block(nn.Module):
    def __init__(self, dim, drop_path=0.0, layer_scale_init=1e-6):
        super().__init__()
        self.dwconv = nn.Conv2d(dim, dim, kernel_size=7, padding=3, groups=dim)
        self.norm = nn.LayerNorm(dim, eps=1e-6)
        self.pwconv1 = nn.Linear(dim, 4 * dim)
        self.act = nn.GELU()
        self.pwconv2 = nn.Linear(4 * dim, dim)
        self.gamma = nn.Parameter(layer_scale_init * torch.ones(dim))
        self.drop_path = DropPath(drop_path) if drop_path > 0 else nn.Identity()

    def forward(self, x):
        shortcut = x
        x = self.dwconv(x)
        x = x.permute(0, 2, 3, 1)  # channels-last for LN + pointwise
        x = self.pwconv2(self.act(self.pwconv1(self.norm(x))))
        x = (self.gamma * x).permute(0, 3, 1, 2)
        return shortcut + self.drop_path(x)


class ConvNeXt(nn.Module):
    """ConvNeXt-T by default (depths 3-3-9-3, dims 96-192-384-768)."""

    @configurable
    def __init__(self, img_size=224, in_chans=3, num_classes=1000,
                 depths=(3, 3, 9, 3), dims=(96, 192, 384, 768),
                 drop_path_rate=0.0, layer_scale_init=1e-6, loss_func=None):
        super().__init__()
        self.downsample_layers = nn.ModuleList()
        stem = nn.Sequential(
            nn.Conv2d(in_chans, dims[0], kernel_size=4, stride=4),
            _ChannelsFirstLN(dims[0]),
        )
        self.downsample_layers.append(stem)
        for i in range(3):
            self.downsample_layers.append(nn.Sequential(
                _ChannelsFirstLN(dims[i]),
                nn.Conv2d(dims[i], dims[i + 1], kernel_size=2, stride=2),
            ))
        self.stages = nn.ModuleList()
        dp_rates = torch.linspace(0, drop_path_rate, sum(depths)).tolist()
        cur = 0
        for i in range(4):
            self.stages.append(nn.Sequential(*[
                _Block(dims[i], dp_rates[cur + j], layer_scale_init)
                for j in range(depths[i])
            ]))
            cur += depths[i]
        self.norm = nn.LayerNorm(dims[-1], eps=1e-6)
        self.head = nn.Linear(dims[-1], num_classes)
        self.loss_func = loss_func or nn.CrossEntropyLoss()
        self.apply(self._init_weights)

    @classmethod
    def from_config(cls, cfg):
        return {k: cfg.get(k) for k in (
            "img_size", "in_chans", "num_classes", "depths", "dims",
            "drop_path_rate", "layer_scale_init",
        ) if cfg.get(k) is not None}

    @staticmethod
    def _init_weights(m):
        if isinstance(m, (nn.Conv2d, nn.Linear)):
            nn.init.trunc_normal_(m.weight, std=0.02)
            if m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward_features(self, images):
        x = images
        for down, stage in zip(self.downsample_layers, self.stages):
            x = stage(down(x))
        return self.norm(x.mean([-2, -1]))  # global average pool

    def forward(self, images, labels=None):
        logits = self.head(self.forward_features(images))
        if labels is not None and self.training:
            return {"losses": self.loss_func(logits, labels)}
        return {"prediction_scores": logits}


class _ChannelsFirstLN(nn.Module):
    def __init__(self, dim, eps=1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))
        self.eps = eps

    def forward(self, x):  # [b, c, h, w]
        u = x.mean(1, keepdim=True)
        s = (x - u).pow(2).mean(1, keepdim=True)
        x = (x - u) / torch.sqrt(s + self.eps)
        return self.weight[:, None, None] * x + self.bias[:, None, None]
