"""Fused scale + mask + softmax (+ dropout) for attention scores.

Replaces the reference's flow._C.fused_scale_tril_softmax_mask_scale and
flow._C.fused_scale_mask_softmax_dropout (reference:
libai/layers/attention.py:221-246).  The backward recomputes the softmax from
the saved input scores and the dropout mask from philox — only the scores and
the output cross HBM.
"""

import torch

from ._ext import draw_seed, ext, use_hip

__all__ = ["fused_scale_mask_softmax"]


class _ScaleMaskSoftmaxFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, scores, pad_mask, scale, p, causal):
        scores = scores.contiguous()
        seed = draw_seed() if p > 0 else 0
        out = ext().softmax_fwd(scores, pad_mask, scale, p, seed, causal)
        ctx.save_for_backward(scores, *(() if pad_mask is None else (pad_mask,)))
        ctx.meta = (scale, p, seed, causal, pad_mask is not None)
        return out

    @staticmethod
    def backward(ctx, dout):
        scale, p, seed, causal, has_mask = ctx.meta
        if has_mask:
            scores, pad_mask = ctx.saved_tensors
        else:
            (scores,) = ctx.saved_tensors
            pad_mask = None
        ds = ext().softmax_bwd(scores, dout.contiguous(), pad_mask, scale, p, seed, causal)
        return ds, None, None, None, None


def _ref(scores, pad_mask, scale, p, causal, training):
    s = scores.float() * scale
    sq, sk = s.shape[-2], s.shape[-1]
    if causal:
        mask = torch.ones(sq, sk, dtype=torch.bool, device=s.device).tril_(sk - sq)
        s = s.masked_fill(~mask, float("-inf"))
    if pad_mask is not None:
        # pad_mask: [B, SQ, SK], 1 = masked (additive -10000, reference semantics)
        s = s - 10000.0 * pad_mask[:, None, :, :].float()
    probs = torch.softmax(s, dim=-1).to(scores.dtype)
    if p > 0 and training:
        probs = torch.nn.functional.dropout(probs, p=p, training=True)
    return probs


def fused_scale_mask_softmax(scores, pad_mask=None, scale=1.0, p=0.0, causal=True,
                             training=True):
    """scores: [B, NH, SQ, SK]; pad_mask: optional [B, SQ, SK] bool/uint8 (1=mask).

    Returns dropped attention probabilities in the input dtype.
    """
    if use_hip(scores) and scores.shape[-1] % 8 == 0 and scores.shape[-1] <= 8192:
        if pad_mask is not None:
            pad_mask = pad_mask.to(torch.uint8).contiguous()
        return _ScaleMaskSoftmaxFn.apply(
            scores, pad_mask, scale, p if training else 0.0, causal
        )
    if scores.is_cuda and training and scores.requires_grad:
        # non-conforming score widths (e.g. ViT's 8x8 patches + cls = 65
        # tokens) run the composed torch path; this is a SHAPE fallback, not
        # an extension fallback — log it once so it cannot hide silently.
        from ..utils.logger import log_first_n
        import logging

        log_first_n(
            logging.WARNING,
            f"fused softmax: SK={scores.shape[-1]} is not a multiple of 8 "
            f"(or > 8192); using the composed torch path for this shape",
            n=1,
        )
    # odd widths (ViT token counts, KV-cache decode steps) -> composed path
    return _ref(scores, pad_mask, scale, p, causal, training)
