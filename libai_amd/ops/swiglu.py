"""Fused SwiGLU: y = silu(gate) * up (reference: the gated-gelu/silu MLPs of
projects/MT5 and projects/Llama)."""

import torch
import torch.nn.functional as F

from ._ext import ext, use_hip

__all__ = ["swiglu"]


class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        x = x.contiguous()
        ctx.save_for_backward(x)
        return ext().swiglu_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        return ext().swiglu_bwd(x, dy.contiguous())


def swiglu(x):
    """x: [..., 2F] (gate | up halves) -> [..., F]."""
    if use_hip(x):
        return _SwiGLUFn.apply(x)
    gate, up = x.chunk(2, dim=-1)
    return F.silu(gate) * up
