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

__all__ = ["flash_attention", "flash_attention_available"]


def flash_attention_available(head_dim, dtype, device, sq, sk, pad_mask):
    return (
        device.type == "cuda"
        and dtype == torch.bfloat16
        and head_dim in (64, 128)
        and pad_mask is None
        and sk % 8 == 0
        and sq == sk  # training self-attention (no KV-cache decode)
    )


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
        B, Sq, H, D = q.shape
        Sk = k.shape[1]

        # [B, H, S, D] views for the batched GEMMs
        qh = q.permute(0, 2, 1, 3)
        kh = k.permute(0, 2, 1, 3)
        vh = v.permute(0, 2, 1, 3)
        doh = do.permute(0, 2, 1, 3)

        s = torch.matmul(qh, kh.transpose(-1, -2))  # [B,H,Sq,Sk] bf16
        p = torch.exp(s.float() * scale - lse.unsqueeze(-1))
        if causal:
            cm = torch.ones(Sq, Sk, dtype=torch.bool, device=q.device).tril_(Sk - Sq)
            p = p.masked_fill(~cm, 0.0)
        p = p.to(q.dtype)
        if p_drop > 0:
            pd = p.clone()
            ext().attn_dropout_apply(pd, Sq, Sk, p_drop, seed)
        else:
            pd = p

        dv = torch.matmul(pd.transpose(-1, -2), doh)  # [B,H,Sk,D]
        dpd = torch.matmul(doh, vh.transpose(-1, -2))  # [B,H,Sq,Sk]
        if p_drop > 0:
            ext().attn_dropout_apply(dpd, Sq, Sk, p_drop, seed)
        # rowsum(dP*P) == rowsum(dO*O) (holds with dropout; see flash-attn)
        Drow = (doh.float() * o.permute(0, 2, 1, 3).float()).sum(-1)  # [B,H,Sq]
        ds = (p.float() * (dpd.float() - Drow.unsqueeze(-1)) * scale).to(q.dtype)
        dq = torch.matmul(ds, kh)  # [B,H,Sq,D]
        dk = torch.matmul(ds.transpose(-1, -2), qh)  # [B,H,Sk,D]

        return (
            dq.permute(0, 2, 1, 3),
            dk.permute(0, 2, 1, 3),
            dv.permute(0, 2, 1, 3),
            None,
            None,
            None,
        )


def flash_attention(q, k, v, scale, p_drop=0.0, causal=True, training=True):
    """q, k, v: [B, S, H, D] bf16 -> O [B, S, H, D]."""
    return _FlashAttnFn.apply(q, k, v, scale, p_drop if training else 0.0, causal)
