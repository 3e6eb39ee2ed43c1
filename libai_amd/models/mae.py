"""Masked Autoencoder (MAE) self-supervised pretraining.

Reference capability: projects/MAE (ViT encoder on visible patches +
lightweight decoder reconstructing masked patches, per-patch normalized
MSE on masked positions) built on the library's layer stack.
"""

import torch
from torch import nn

from ..config import configurable
from ..layers import LayerNorm, PatchEmbedding, TransformerLayer
from .utils.weight_init import init_method_normal

__all__ = ["MAEForPreTraining"]


def _sincos_pos_embed(n, dim, device=None):
    """Fixed 1D sin-cos position embedding [n, dim] (cls row excluded)."""
    import math

    pos = torch.arange(n, dtype=torch.float32, device=device)[:, None]
    div = torch.exp(torch.arange(0, dim, 2, dtype=torch.float32, device=device)
                    * (-math.log(10000.0) / dim))
    pe = torch.zeros(n, dim, device=device)
    pe[:, 0::2] = torch.sin(pos * div)
    pe[:, 1::2] = torch.cos(pos * div)
    return pe


class MAEForPreTraining(nn.Module):
    @configurable
    def __init__(
        self,
        img_size=224,
        patch_size=16,
        in_chans=3,
        embed_dim=768,
        depth=12,
        num_heads=12,
        decoder_embed_dim=512,
        decoder_depth=8,
        decoder_num_heads=16,
        mask_ratio=0.75,
        norm_pix_loss=True,
        initializer_range=0.02,
    ):
        super().__init__()
        init_method = init_method_normal(initializer_range)
        self.patch_size = patch_size
        self.in_chans = in_chans
        self.mask_ratio = mask_ratio
        self.norm_pix_loss = norm_pix_loss

        # encoder (runs on VISIBLE patches only — the MAE speed trick)
        self.patch_embed = PatchEmbedding(img_size, patch_size, in_chans,
                                          embed_dim, layer_idx=0)
        n_patches = self.patch_embed.num_patches
        self.cls_token = nn.Parameter(torch.zeros(1, 1, embed_dim))
        self.register_buffer(
            "pos_embed",
            torch.cat([torch.zeros(1, embed_dim),
                       _sincos_pos_embed(n_patches, embed_dim)])[None],
            persistent=False,
        )
        self.blocks = nn.ModuleList([
            TransformerLayer(embed_dim, 4 * embed_dim, num_heads,
                             init_method=init_method, layer_idx=i)
            for i in range(depth)
        ])
        self.norm = LayerNorm(embed_dim, layer_idx=-1)

        # decoder (small, reconstructs every patch)
        self.decoder_embed = nn.Linear(embed_dim, decoder_embed_dim)
        self.mask_token = nn.Parameter(torch.zeros(1, 1, decoder_embed_dim))
        self.register_buffer(
            "decoder_pos_embed",
            torch.cat([torch.zeros(1, decoder_embed_dim),
                       _sincos_pos_embed(n_patches, decoder_embed_dim)])[None],
            persistent=False,
        )
        self.decoder_blocks = nn.ModuleList([
            TransformerLayer(decoder_embed_dim, 4 * decoder_embed_dim,
                             decoder_num_heads, init_method=init_method,
                             layer_idx=-1)
            for _ in range(decoder_depth)
        ])
        self.decoder_norm = LayerNorm(decoder_embed_dim, layer_idx=-1)
        self.decoder_pred = nn.Linear(decoder_embed_dim,
                                      patch_size * patch_size * in_chans)
        nn.init.normal_(self.cls_token, std=0.02)
        nn.init.normal_(self.mask_token, std=0.02)

    @classmethod
    def from_config(cls, cfg):
        return {k: cfg.get(k) for k in (
            "img_size", "patch_size", "in_chans", "embed_dim", "depth",
            "num_heads", "decoder_embed_dim", "decoder_depth",
            "decoder_num_heads", "mask_ratio", "norm_pix_loss",
        ) if cfg.get(k) is not None}

    # -- patch pixel targets -------------------------------------------------

    def patchify(self, imgs):
        p = self.patch_size
        B, C, H, W = imgs.shape
        x = imgs.reshape(B, C, H // p, p, W // p, p)
        x = x.permute(0, 2, 4, 3, 5, 1).reshape(B, (H // p) * (W // p), p * p * C)
        return x

    def random_masking(self, x):
        """Per-sample random shuffle; keep (1 - mask_ratio) of patches."""
        B, N, D = x.shape
        keep = max(1, int(N * (1 - self.mask_ratio)))
        noise = torch.rand(B, N, device=x.device)
        shuffle = torch.argsort(noise, dim=1)
        restore = torch.argsort(shuffle, dim=1)
        kept = shuffle[:, :keep]
        x_vis = torch.gather(x, 1, kept[..., None].expand(-1, -1, D))
        mask = torch.ones(B, N, device=x.device)
        mask[:, :keep] = 0
        mask = torch.gather(mask, 1, restore)  # 1 = masked (to reconstruct)
        return x_vis, mask, restore

    def forward(self, images, labels=None):
        x = self.patch_embed(images)  # [B, N, D]
        x = x + self.pos_embed[:, 1:].to(x.dtype)
        x_vis, mask, restore = self.random_masking(x)
        cls = (self.cls_token + self.pos_embed[:, :1].to(x.dtype)).expand(
            x.shape[0], -1, -1)
        h = torch.cat([cls, x_vis], dim=1)
        for blk in self.blocks:
            h = blk(h)
        h = self.norm(h)

        # decoder: re-insert mask tokens at their original positions
        h = self.decoder_embed(h)
        B, N = mask.shape
        n_mask = N - (h.shape[1] - 1)
        mask_tokens = self.mask_token.expand(B, n_mask, -1).to(h.dtype)
        full = torch.cat([h[:, 1:], mask_tokens], dim=1)
        full = torch.gather(
            full, 1, restore[..., None].expand(-1, -1, full.shape[-1]))
        full = torch.cat([h[:, :1], full], dim=1)
        full = full + self.decoder_pos_embed.to(full.dtype)
        for blk in self.decoder_blocks:
            full = blk(full)
        pred = self.decoder_pred(self.decoder_norm(full))[:, 1:]  # [B, N, p*p*C]

        target = self.patchify(images).to(pred.dtype)
        if self.norm_pix_loss:
            mean = target.mean(dim=-1, keepdim=True)
            var = target.var(dim=-1, keepdim=True)
            target = (target - mean) / (var + 1e-6).sqrt()
        loss = ((pred - target) ** 2).mean(dim=-1)  # per-patch MSE
        loss = (loss * mask).sum() / mask.sum().clamp(min=1.0)
        return {"mae_loss": loss}

    def set_activation_checkpoint(self, enabled=True):
        pass  # encoder already runs on 25% of tokens; no-op for parity
