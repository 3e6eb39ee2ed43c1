"""LayerNorm / RMSNorm functional ops: HIP kernels on GPU, torch reference on CPU.

Replaces the reference's flow._C.layer_norm_affine / flow._C.rms_norm
(reference: libai/layers/layer_norm.py:78-131).
"""

import torch
import torch.nn.functional as F

from ._ext import ext, use_hip

__all__ = ["layer_norm", "rms_norm"]


class _NormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps, rms):
        x = x.contiguous()
        y, mean, rstd = ext().ln_fwd(x, weight.contiguous(),
                                     bias.contiguous() if bias is not None else None,
                                     rms, eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        ctx.rms = rms
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        dx, dgamma, dbeta = ext().ln_bwd(
            dy.contiguous(), x, weight, mean, rstd, ctx.rms, ctx.has_bias
        )
        return dx, dgamma, (dbeta if ctx.has_bias else None), None, None


def layer_norm(x, weight, bias, eps=1e-5):
    if use_hip(x):
        return _NormFn.apply(x, weight, bias, eps, False)
    return F.layer_norm(x, (weight.numel(),), weight, bias, eps)


def _rms_norm_ref(x, weight, eps):
    dt = x.dtype
    xf = x.float()
    rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * rstd).to(dt) * weight


def rms_norm(x, weight, eps=1e-5):
    if use_hip(x):
        return _NormFn.apply(x, weight, None, eps, True)
    return _rms_norm_ref(x, weight, eps)
