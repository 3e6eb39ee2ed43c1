"""Swin Transformer (reference: libai/models/swin_transformer.py, 772 LoC).

Windowed self-attention with relative position bias, shifted windows,
patch merging between stages.  The reference runs Swin data-parallel only
(Model_Zoo.md); this implementation follows suit (plain nn.Linear inside
window attention, DP via the trainer).
"""

import torch
import torch.nn.functional as F
from torch import nn

from ..config import configurable
from ..layers import DropPath, build_activation
from .utils.weight_init import init_method_normal

__all__ = ["SwinTransformer"]


def window_partition(x, ws):
    B, H, W, C = x.shape
    x = x.view(B, H // ws, ws, W // ws, ws, C)
    return x.permute(0, 1, 3, 2, 4, 5).reshape(-1, ws * ws, C)


def window_reverse(windows, ws, H, W):
    B = windows.shape[0] // (H * W // ws // ws)
    x = windows.view(B, H // ws, W // ws, ws, ws, -1)
    return x.permute(0, 1, 3, 2, 4, 5).reshape(B, H, W, -1)


class WindowAttention(nn.Module):
    def __init__(self, dim, window_size, num_heads, qkv_bias=True, attn_drop=0.0,
                 proj_drop=0.0):
        super().__init__()
        self.dim = dim
        self.window_size = window_size
        self.num_heads = num_heads
        self.scale = (dim // num_heads) ** -0.5

        self.relative_position_bias_table = nn.Parameter(
            torch.zeros((2 * window_size - 1) ** 2, num_heads)
        )
        coords = torch.stack(
            torch.meshgrid(
                torch.arange(window_size), torch.arange(window_size), indexing="ij"
            )
        ).flatten(1)
        rel = coords[:, :, None] - coords[:, None, :]
        rel = rel.permute(1, 2, 0).contiguous()
        rel[:, :, 0] += window_size - 1
        rel[:, :, 1] += window_size - 1
        rel[:, :, 0] *= 2 * window_size - 1
        self.register_buffer("relative_position_index", rel.sum(-1), persistent=False)
        nn.init.trunc_normal_(self.relative_position_bias_table, std=0.02)

        self.qkv = nn.Linear(dim, dim * 3, bias=qkv_bias)
        self.attn_drop = nn.Dropout(attn_drop)
        self.proj = nn.Linear(dim, dim)
        self.proj_drop = nn.Dropout(proj_drop)

    def forward(self, x, mask=None):
        B_, N, C = x.shape
        qkv = (
            self.qkv(x)
            .reshape(B_, N, 3, self.num_heads, C // self.num_heads)
            .permute(2, 0, 3, 1, 4)
        )
        q, k, v = qkv[0], qkv[1], qkv[2]
        attn = (q * self.scale) @ k.transpose(-2, -1)
        bias = self.relative_position_bias_table[
            self.relative_position_index.view(-1)
        ].view(N, N, -1).permute(2, 0, 1)
        attn = attn + bias.unsqueeze(0).to(attn.dtype)
        if mask is not None:
            nW = mask.shape[0]
            attn = attn.view(B_ // nW, nW, self.num_heads, N, N) + mask.unsqueeze(
                1
            ).unsqueeze(0).to(attn.dtype)
            attn = attn.view(-1, self.num_heads, N, N)
        attn = torch.softmax(attn.float(), dim=-1).to(x.dtype)
        attn = self.attn_drop(attn)
        x = (attn @ v).transpose(1, 2).reshape(B_, N, C)
        return self.proj_drop(self.proj(x))


class SwinBlock(nn.Module):
    def __init__(self, dim, input_resolution, num_heads, window_size=7, shift_size=0,
                 mlp_ratio=4.0, qkv_bias=True, drop=0.0, attn_drop=0.0,
                 drop_path=0.0, *, layer_idx=0):
        super().__init__()
        self.dim = dim
        self.input_resolution = input_resolution
        self.window_size = min(window_size, min(input_resolution))
        self.shift_size = 0 if self.window_size == min(input_resolution) else shift_size
        self.layer_idx = layer_idx

        self.norm1 = nn.LayerNorm(dim)
        self.attn = WindowAttention(dim, self.window_size, num_heads, qkv_bias,
                                    attn_drop, drop)
        self.drop_path = DropPath(drop_path)
        self.norm2 = nn.LayerNorm(dim)
        hidden = int(dim * mlp_ratio)
        self.mlp = nn.Sequential(
            nn.Linear(dim, hidden), build_activation("gelu"), nn.Dropout(drop),
            nn.Linear(hidden, dim), nn.Dropout(drop),
        )

        if self.shift_size > 0:
            H, W = input_resolution
            img_mask = torch.zeros(1, H, W, 1)
            slices = (
                slice(0, -self.window_size),
                slice(-self.window_size, -self.shift_size),
                slice(-self.shift_size, None),
            )
            cnt = 0
            for h in slices:
                for w in slices:
                    img_mask[:, h, w, :] = cnt
                    cnt += 1
            mw = window_partition(img_mask, self.window_size).squeeze(-1)
            attn_mask = mw.unsqueeze(1) - mw.unsqueeze(2)
            attn_mask = attn_mask.masked_fill(attn_mask != 0, -100.0)
            self.register_buffer("attn_mask", attn_mask, persistent=False)
        else:
            self.attn_mask = None

    def forward(self, x):
        H, W = self.input_resolution
        B, L, C = x.shape
        shortcut = x
        x = self.norm1(x).view(B, H, W, C)
        if self.shift_size > 0:
            x = torch.roll(x, shifts=(-self.shift_size, -self.shift_size), dims=(1, 2))
        windows = window_partition(x, self.window_size)
        attn_windows = self.attn(windows, mask=self.attn_mask)
        x = window_reverse(attn_windows, self.window_size, H, W)
        if self.shift_size > 0:
            x = torch.roll(x, shifts=(self.shift_size, self.shift_size), dims=(1, 2))
        x = shortcut + self.drop_path(x.view(B, L, C))
        x = x + self.drop_path(self.mlp(self.norm2(x)))
        return x


class PatchMerging(nn.Module):
    def __init__(self, input_resolution, dim):
        super().__init__()
        self.input_resolution = input_resolution
        self.dim = dim
        self.reduction = nn.Linear(4 * dim, 2 * dim, bias=False)
        self.norm = nn.LayerNorm(4 * dim)

    def forward(self, x):
        H, W = self.input_resolution
        B, L, C = x.shape
        x = x.view(B, H, W, C)
        x = torch.cat(
            [x[:, 0::2, 0::2], x[:, 1::2, 0::2], x[:, 0::2, 1::2], x[:, 1::2, 1::2]],
            dim=-1,
        ).view(B, -1, 4 * C)
        return self.reduction(self.norm(x))


class SwinTransformer(nn.Module):
    @configurable
    def __init__(self, img_size=224, patch_size=4, in_chans=3, num_classes=1000,
                 embed_dim=96, depths=(2, 2, 6, 2), num_heads=(3, 6, 12, 24),
                 window_size=7, mlp_ratio=4.0, qkv_bias=True, drop_rate=0.0,
                 attn_drop_rate=0.0, drop_path_rate=0.1, loss_func=None):
        super().__init__()
        self.num_classes = num_classes
        self.num_layers = len(depths)
        self.patch_embed = nn.Conv2d(in_chans, embed_dim, kernel_size=patch_size,
                                     stride=patch_size)
        self.patch_norm = nn.LayerNorm(embed_dim)
        patches_res = (img_size // patch_size, img_size // patch_size)
        self.pos_drop = nn.Dropout(drop_rate)

        dpr = torch.linspace(0, drop_path_rate, sum(depths)).tolist()
        self.layers = nn.ModuleList()
        dim = embed_dim
        res = patches_res
        li = 0
        for i, depth in enumerate(depths):
            stage = nn.ModuleList(
                [
                    SwinBlock(
                        dim, res, num_heads[i], window_size,
                        shift_size=0 if (j % 2 == 0) else window_size // 2,
                        mlp_ratio=mlp_ratio, qkv_bias=qkv_bias, drop=drop_rate,
                        attn_drop=attn_drop_rate, drop_path=dpr[li + j],
                        layer_idx=li + j,
                    )
                    for j in range(depth)
                ]
            )
            li += depth
            merge = PatchMerging(res, dim) if i < len(depths) - 1 else None
            self.layers.append(nn.ModuleList([stage, merge] if merge else [stage]))
            if merge is not None:
                dim *= 2
                res = (res[0] // 2, res[1] // 2)
        self.norm = nn.LayerNorm(dim)
        self.head = nn.Linear(dim, num_classes)
        self.loss_func = nn.CrossEntropyLoss() if loss_func is None else loss_func

    @classmethod
    def from_config(cls, cfg):
        return {
            "img_size": cfg.get("img_size", 224),
            "patch_size": cfg.get("patch_size", 4),
            "in_chans": cfg.get("in_chans", 3),
            "num_classes": cfg.get("num_classes", 1000),
            "embed_dim": cfg.get("embed_dim", 96),
            "depths": cfg.get("depths", (2, 2, 6, 2)),
            "num_heads": cfg.get("num_heads", (3, 6, 12, 24)),
            "window_size": cfg.get("window_size", 7),
            "mlp_ratio": cfg.get("mlp_ratio", 4.0),
            "qkv_bias": cfg.get("qkv_bias", True),
            "drop_rate": cfg.get("drop_rate", 0.0),
            "attn_drop_rate": cfg.get("attn_drop_rate", 0.0),
            "drop_path_rate": cfg.get("drop_path_rate", 0.1),
            "loss_func": cfg.get("loss_func", None),
        }

    def forward_features(self, x):
        x = self.patch_embed(x).flatten(2).transpose(1, 2)
        x = self.pos_drop(self.patch_norm(x))
        for stage in self.layers:
            for block in stage[0]:
                x = block(x)
            if len(stage) > 1 and stage[1] is not None:
                x = stage[1](x)
        return self.norm(x).mean(dim=1)

    def forward(self, images, labels=None):
        feats = self.forward_features(images)
        logits = self.head(feats)
        if labels is not None and self.training:
            return {"losses": self.loss_func(logits.float(), labels)}
        return {"prediction_scores": logits}
