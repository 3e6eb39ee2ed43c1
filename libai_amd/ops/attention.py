"""Fused flash-style attention (bf16, MFMA) with recompute backward.

Forward: one HIP kernel (csrc/kernels/flash_attn.hip) — the S x S score
matrix never touches HBM; O and the per-row logsumexp are saved.

Backward (v1): recompute composition — S = QK^T (rocBLAS), P = exp(S*scale -
lse) with the philox dropout mask regenerated in-kernel, then the four
gradient GEMMs.  A fully-fused HIP backward is the planned v2.

Replaces the reference's K2-K5 chain (reference
libai/layers/attention.py:211-253).
"""

import torch

from ._ext import draw_seed, ext

__all__ = ["flash_attention", "flash_attention_qkv", "flash_attention_available"]


def flash_attention_available(head_dim, dtype, device, sq, sk, pad_mask):
    return (
        device.type == "cuda"
        and dtype == torch.bfloat16
        and head_dim in (64, 128)
        and pad_mask is None
        and sk % 8 == 0
        and sq == sk  # training self-attention (no KV-cache decode)
    )


class _FlashAttnQKVFn(torch.autograd.Function):
    """Packed-qkv variant: takes the fused [B, S, H, 3, D] projection buffer
    directly.  The backward writes dQ/dK/dV straight into one packed dqkv
    allocation (the HIP kernels take strides), so autograd never materializes
    three view-grads and scatter-adds them into a zeroed buffer -- that glue
    measured ~5 ms/step on GPT-2 345M (three 50M-element passes per layer).
    """

    @staticmethod
    def forward(ctx, qkv5, scale, p_drop, causal):
        q = qkv5[..., 0, :]
        k = qkv5[..., 1, :]
        v = qkv5[..., 2, :]
        seed = draw_seed() if p_drop > 0 else 0
        o, lse = ext().flash_fwd(q, k, v, scale, p_drop, seed, causal)
        ctx.save_for_backward(qkv5, o, lse)
        ctx.meta = (scale, p_drop, seed, causal)
        return o

    @staticmethod
    def backward(ctx, do):
        qkv5, o, lse = ctx.saved_tensors
        scale, p_drop, seed, causal = ctx.meta
        dqkv = torch.empty_like(qkv5)
        ext().flash_bwd(
            qkv5[..., 0, :], qkv5[..., 1, :], qkv5[..., 2, :], o, do.contiguous(),
            lse, scale, p_drop, seed, causal, dqkv,
        )
        return dqkv, None, None, None


def flash_attention_qkv(qkv5, scale, p_drop=0.0, causal=True, training=True):
    """qkv5: packed [B, S, H, 3, D] bf16 -> O [B, S, H, D] (zero-copy in/out)."""
    return _FlashAttnQKVFn.apply(qkv5, scale, p_drop if training else 0.0, causal)


class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale, p_drop, causal):
        # q, k, v: [B, S, H, D] (may be strided views of the fused qkv buffer)
        seed = draw_seed() if p_drop > 0 else 0
        o, lse = ext().flash_fwd(q, k, v, scale, p_drop, seed, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.meta = (scale, p_drop, seed, causal)
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        scale, p_drop, seed, causal = ctx.meta
        dq, dk, dv = ext().flash_bwd(
            q, k, v, o, do.contiguous(), lse, scale, p_drop, seed, causal, None
        )
        return dq, dk, dv, None, None, None


def flash_attention(q, k, v, scale, p_drop=0.0, causal=True, training=True):
    """q, k, v: [B, S, H, D] bf16 -> O [B, S, H, D]."""
    return _FlashAttnFn.apply(q, k, v, scale, p_drop if training else 0.0, causal)
