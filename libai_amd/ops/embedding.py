"""Fused embedding gather + scatter-add backward (K10).

Replaces torch's F.embedding on the GPU hot path (reference site:
libai/layers/embedding.py:85,164): the forward folds the vocab-parallel
OOV-zeroing in-kernel, the backward is one fp32 atomic scatter + cast
instead of torch's sort/reduce composition (~4% of the GPT-2 345M step in
the round-1 profile).
"""

import torch

from ._ext import ext, use_hip

__all__ = ["fused_embedding", "fused_embedding_available"]


def fused_embedding_available(weight):
    return (
        use_hip(weight)
        and weight.dtype in (torch.bfloat16, torch.float32)
        and (weight.shape[1] * weight.element_size()) % 16 == 0
    )


class _EmbeddingFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, ids, weight, vocab_start, padding_idx):
        ids = ids.contiguous()
        ctx.save_for_backward(ids)
        ctx.meta = (weight.shape[0], vocab_start, padding_idx, weight.dtype)
        return ext().embedding_fwd(ids, weight, vocab_start)

    @staticmethod
    def backward(ctx, dout):
        (ids,) = ctx.saved_tensors
        vocab_local, vocab_start, padding_idx, dtype = ctx.meta
        dw = ext().embedding_bwd(ids, dout, vocab_local, vocab_start,
                                 padding_idx, dtype)
        return None, dw, None, None


def fused_embedding(ids, weight, vocab_start=0, padding_idx=None):
    """ids [*] int64, weight [vocab_local, H] -> [*, H].

    ids outside [vocab_start, vocab_start + vocab_local) yield zero rows
    (the vocab-parallel partial sum, all-reduced by the caller)."""
    return _EmbeddingFn.apply(ids, weight, vocab_start,
                              -1 if padding_idx is None else int(padding_idx))
