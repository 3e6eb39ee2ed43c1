"""Swin Transformer V2 (reference: libai/models/swin_transformer_v2.py).

V2 deltas over V1: scaled-cosine attention with a learnable per-head logit
scale, log-spaced continuous relative position bias from a small MLP, and
post-normalization (norm after attention/MLP residual branches).
"""

import math

import torch
import torch.nn.functional as F
from torch import nn

from ..config import configurable
from .swin_transformer import (
    PatchMerging,
    SwinTransformer,
    window_partition,
    window_reverse,
)
from ..layers import DropPath, build_activation

__all__ = ["SwinTransformerV2"]


class WindowAttentionV2(nn.Module):
    def __init__(self, dim, window_size, num_heads, qkv_bias=True, attn_drop=0.0,
                 proj_drop=0.0):
        super().__init__()
        self.dim = dim
        self.window_size = window_size
        self.num_heads = num_heads
        self.logit_scale = nn.Parameter(
            torch.log(10 * torch.ones(num_heads, 1, 1))
        )
        # continuous position bias MLP over log-spaced coords
        self.cpb_mlp = nn.Sequential(
            nn.Linear(2, 512, bias=True), nn.ReLU(inplace=True),
            nn.Linear(512, num_heads, bias=False),
        )
        coords = torch.stack(
            torch.meshgrid(
                torch.arange(-(window_size - 1), window_size, dtype=torch.float32),
                torch.arange(-(window_size - 1), window_size, dtype=torch.float32),
                indexing="ij",
            )
        ).permute(1, 2, 0)  # [2w-1, 2w-1, 2]
        coords = coords / (window_size - 1) * 8
        coords = torch.sign(coords) * torch.log2(coords.abs() + 1.0) / math.log2(8)
        self.register_buffer("relative_coords_table", coords.unsqueeze(0),
                             persistent=False)

        c = torch.stack(
            torch.meshgrid(torch.arange(window_size), torch.arange(window_size),
                           indexing="ij")
        ).flatten(1)
        rel = (c[:, :, None] - c[:, None, :]).permute(1, 2, 0).contiguous()
        rel[:, :, 0] += window_size - 1
        rel[:, :, 1] += window_size - 1
        rel[:, :, 0] *= 2 * window_size - 1
        self.register_buffer("relative_position_index", rel.sum(-1), persistent=False)

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
        # cosine attention with clamped learnable scale
        attn = F.normalize(q.float(), dim=-1) @ F.normalize(k.float(), dim=-1
                                                            ).transpose(-2, -1)
        scale = torch.clamp(self.logit_scale,
                            max=math.log(1.0 / 0.01)).exp().to(attn.dtype)
        attn = attn * scale
        bias_table = self.cpb_mlp(self.relative_coords_table).view(-1, self.num_heads)
        bias = bias_table[self.relative_position_index.view(-1)].view(
            N, N, -1
        ).permute(2, 0, 1)
        attn = attn + (16 * torch.sigmoid(bias)).unsqueeze(0)
        if mask is not None:
            nW = mask.shape[0]
            attn = attn.view(B_ // nW, nW, self.num_heads, N, N) + mask.unsqueeze(
                1
            ).unsqueeze(0)
            attn = attn.view(-1, self.num_heads, N, N)
        attn = torch.softmax(attn, dim=-1).to(x.dtype)
        attn = self.attn_drop(attn)
        x = (attn @ v).transpose(1, 2).reshape(B_, N, C)
        return self.proj_drop(self.proj(x))


class SwinBlockV2(nn.Module):
    """Post-norm Swin block with cosine window attention."""

    def __init__(self, dim, input_resolution, num_heads, window_size=7, shift_size=0,
                 mlp_ratio=4.0, qkv_bias=True, drop=0.0, attn_drop=0.0,
                 drop_path=0.0, *, layer_idx=0):
        super().__init__()
        self.input_resolution = input_resolution
        self.window_size = min(window_size, min(input_resolution))
        self.shift_size = 0 if self.window_size == min(input_resolution) else shift_size
        self.layer_idx = layer_idx
        self.attn = WindowAttentionV2(dim, self.window_size, num_heads, qkv_bias,
                                      attn_drop, drop)
        self.norm1 = nn.LayerNorm(dim)  # applied AFTER attention (v2 post-norm)
        self.drop_path = DropPath(drop_path)
        hidden = int(dim * mlp_ratio)
        self.mlp = nn.Sequential(
            nn.Linear(dim, hidden), build_activation("gelu"), nn.Dropout(drop),
            nn.Linear(hidden, dim), nn.Dropout(drop),
        )
        self.norm2 = nn.LayerNorm(dim)

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
        xs = x.view(B, H, W, C)
        if self.shift_size > 0:
            xs = torch.roll(xs, shifts=(-self.shift_size, -self.shift_size),
                            dims=(1, 2))
        windows = window_partition(xs, self.window_size)
        attn_windows = self.attn(windows, mask=self.attn_mask)
        xs = window_reverse(attn_windows, self.window_size, H, W)
        if self.shift_size > 0:
            xs = torch.roll(xs, shifts=(self.shift_size, self.shift_size), dims=(1, 2))
        x = shortcut + self.drop_path(self.norm1(xs.reshape(B, L, C)))
        x = x + self.drop_path(self.norm2(self.mlp(x)))
        return x


class SwinTransformerV2(SwinTransformer):
    @configurable
    def __init__(self, **kwargs):
        super(SwinTransformerV2, self).__init__(**kwargs)
        # rebuild the stages with V2 blocks (same geometry as V1 construction)
        depths = kwargs.get("depths", (2, 2, 6, 2))
        num_heads = kwargs.get("num_heads", (3, 6, 12, 24))
        embed_dim = kwargs.get("embed_dim", 96)
        img_size = kwargs.get("img_size", 224)
        patch_size = kwargs.get("patch_size", 4)
        window_size = kwargs.get("window_size", 7)
        mlp_ratio = kwargs.get("mlp_ratio", 4.0)
        qkv_bias = kwargs.get("qkv_bias", True)
        drop_rate = kwargs.get("drop_rate", 0.0)
        attn_drop_rate = kwargs.get("attn_drop_rate", 0.0)
        drop_path_rate = kwargs.get("drop_path_rate", 0.1)

        dpr = torch.linspace(0, drop_path_rate, sum(depths)).tolist()
        self.layers = nn.ModuleList()
        dim = embed_dim
        res = (img_size // patch_size, img_size // patch_size)
        li = 0
        for i, depth in enumerate(depths):
            stage = nn.ModuleList(
                [
                    SwinBlockV2(
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

    @classmethod
    def from_config(cls, cfg):
        return SwinTransformer.from_config(cfg)
