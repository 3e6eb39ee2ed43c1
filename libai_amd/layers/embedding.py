"""Embedding layers: replicated, vocab-parallel, sinusoidal, patch.

Reference: libai/layers/embedding.py:66-291.  The vocab-parallel variant
shards the table along the vocab dim; out-of-shard ids produce zero rows and
the partial sums are all-reduced over TP (C3 in SURVEY.md §2.4, reference
embedding.py:156-168).
"""

import math

import torch
import torch.nn.functional as F
from torch import nn

from ..parallel.comm import reduce_from_tensor_parallel_region
from ..utils import distributed as du
from .linear import init_tp_shard_

__all__ = ["Embedding", "VocabEmbedding", "SinePositionalEmbedding", "PatchEmbedding"]


class Embedding(nn.Module):
    """Replicated embedding (positions, token types...)."""

    def __init__(self, num_embeddings, embedding_dim, padding_idx=None,
                 init_method=nn.init.xavier_normal_, *, layer_idx=0, dtype=None):
        super().__init__()
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        self.padding_idx = padding_idx
        self.layer_idx = layer_idx
        dtype = dtype or torch.get_default_dtype()
        self.weight = nn.Parameter(torch.empty(num_embeddings, embedding_dim, dtype=dtype))
        init_method(self.weight.data)
        if padding_idx is not None:
            with torch.no_grad():
                self.weight[padding_idx].fill_(0)

    def forward(self, ids):
        from ..ops.embedding import fused_embedding, fused_embedding_available

        if fused_embedding_available(self.weight):
            return fused_embedding(ids, self.weight, 0, self.padding_idx)
        return F.embedding(ids, self.weight, padding_idx=self.padding_idx)


class VocabEmbedding(nn.Module):
    """Vocab-parallel embedding: weight rows [vocab/tp, hidden] per rank."""

    def __init__(self, num_embeddings, embedding_dim, padding_idx=None,
                 init_method=nn.init.xavier_normal_, *, layer_idx=0, dtype=None):
        super().__init__()
        dutil = du.get_dist_util()
        tp, tpr = dutil.tensor_parallel_size, dutil.tensor_parallel_rank
        assert num_embeddings % tp == 0, (num_embeddings, tp)
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        self.padding_idx = padding_idx
        self.layer_idx = layer_idx
        self.vocab_local = num_embeddings // tp
        self.vocab_start = tpr * self.vocab_local
        dtype = dtype or torch.get_default_dtype()
        self.weight = nn.Parameter(
            torch.empty(self.vocab_local, embedding_dim, dtype=dtype)
        )
        self.weight.tensor_parallel = True
        self.weight.tp_shard_dim = 0
        init_tp_shard_(self.weight, (num_embeddings, embedding_dim), init_method, 0)

    def forward(self, ids):
        from ..ops.embedding import fused_embedding, fused_embedding_available

        tp = du.get_dist_util().tensor_parallel_size
        if fused_embedding_available(self.weight):
            # OOV ids -> zero rows in-kernel (no sub/clamp/mask-mul passes)
            out = fused_embedding(ids, self.weight, self.vocab_start,
                                  self.padding_idx)
            return reduce_from_tensor_parallel_region(out) if tp > 1 else out
        if tp == 1:
            return F.embedding(ids, self.weight, padding_idx=self.padding_idx)
        local = ids - self.vocab_start
        in_shard = (local >= 0) & (local < self.vocab_local)
        local = local.clamp(0, self.vocab_local - 1)
        out = F.embedding(local, self.weight)
        out = out * in_shard.unsqueeze(-1).to(out.dtype)
        return reduce_from_tensor_parallel_region(out)


class SinePositionalEmbedding(nn.Module):
    """Fixed sinusoidal positions (reference: embedding.py:186-234)."""

    def __init__(self, num_embeddings, embedding_dim, *, layer_idx=0):
        super().__init__()
        self.layer_idx = layer_idx
        pe = torch.zeros(num_embeddings, embedding_dim)
        position = torch.arange(0, num_embeddings, dtype=torch.float).unsqueeze(1)
        div = torch.exp(
            torch.arange(0, embedding_dim, 2).float()
            * (-math.log(10000.0) / embedding_dim)
        )
        pe[:, 0::2] = torch.sin(position * div)
        pe[:, 1::2] = torch.cos(position * div)
        self.register_buffer("pe", pe, persistent=False)

    def forward(self, positions):
        return F.embedding(positions, self.pe.to(positions.device))


class PatchEmbedding(nn.Module):
    """ViT patchify: Conv2d(k=p, s=p) + flatten (reference: embedding.py:237-291)."""

    def __init__(self, img_size=224, patch_size=16, in_chans=3, embed_dim=768,
                 norm_layer=None, flatten=True, *, layer_idx=0, dtype=None):
        super().__init__()
        img_size = (img_size, img_size) if isinstance(img_size, int) else tuple(img_size)
        patch_size = (
            (patch_size, patch_size) if isinstance(patch_size, int) else tuple(patch_size)
        )
        self.img_size = img_size
        self.patch_size = patch_size
        self.grid_size = (img_size[0] // patch_size[0], img_size[1] // patch_size[1])
        self.num_patches = self.grid_size[0] * self.grid_size[1]
        self.flatten = flatten
        self.layer_idx = layer_idx
        self.proj = nn.Conv2d(in_chans, embed_dim, kernel_size=patch_size,
                              stride=patch_size)
        self.norm = norm_layer(embed_dim) if norm_layer else nn.Identity()

    def forward(self, x):
        B, C, H, W = x.shape
        assert H == self.img_size[0] and W == self.img_size[1], (
            f"input {H}x{W} doesn't match model {self.img_size}"
        )
        x = self.proj(x)
        if self.flatten:
            x = x.flatten(2).transpose(1, 2)  # [B, N, C]
        return self.norm(x)
