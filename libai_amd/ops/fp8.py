"""Opt-in fp8 (e4m3) forward GEMMs — transformer-engine-style mixed
precision: forward matmuls run on the gfx950 fp8 MFMA pipe (dense peak
~5 PFLOP/s, 2x bf16; measured 1.5-2x at the bench shapes,
profiles/gemm_roofline.md), backward stays bf16.

Note: under activation checkpointing the recompute pass quantizes again
with the (slightly advanced) delayed scale, so recomputed activations can
differ from the original forward by one scale step — bounded and standard
for delayed-scaling fp8, but part of why this stays experimental.

Per-tensor DYNAMIC scaling: each forward quantizes activations and weights
with amax/448 scales and lets hipBLASLt dequantize in the epilogue
(`torch._scaled_mm`).  The backward uses the saved bf16 tensors, so
gradient numerics are identical to the bf16 path; only the forward
activations carry the ~2^-3-mantissa rounding.

Enable with ``train.fp8 = dict(enabled=True)`` (or
``libai_amd.ops.fp8.set_fp8_gemms(True)``); every Linear1D GEMM whose
shapes divide 16 routes through here.  The HEADLINE bench stays bf16 —
this is an experimental option beyond the reference's fp16/bf16 modes.
Measured end-to-end at GPT-2 345M it is ~3% slower than bf16 (the GEMMs
win, the per-site bookkeeping costs; profiles/gemm_roofline.md has the
full iterated story) — it is aimed at larger hidden sizes.
"""

import torch

__all__ = ["fp8_available", "set_fp8_gemms", "fp8_gemms_enabled",
           "fp8_eligible", "fp8_linear", "DelayedScale"]

_E4M3_MAX = 448.0
_state = {"enabled": False}
_avail = None


def fp8_available():
    """Probe once: e4m3fn dtype + a working hipBLASLt _scaled_mm path."""
    global _avail
    if _avail is None:
        _avail = False
        if torch.cuda.is_available() and hasattr(torch, "float8_e4m3fn"):
            try:
                a = torch.randn(16, 16, device="cuda").to(torch.float8_e4m3fn)
                s = torch.tensor(1.0, device="cuda")
                torch._scaled_mm(a, a.t(), scale_a=s, scale_b=s,
                                 out_dtype=torch.bfloat16)
                _avail = True
            except Exception:
                _avail = False
    return _avail


def set_fp8_gemms(enabled):
    _state["enabled"] = bool(enabled)


def fp8_gemms_enabled():
    return _state["enabled"]


def fp8_eligible(x, w):
    return (
        _state["enabled"]
        and x.is_cuda
        and x.dtype == torch.bfloat16
        and w.dtype == torch.bfloat16
        and w.shape[0] % 16 == 0
        and w.shape[1] % 16 == 0
        and fp8_available()
    )


def _quant(t):
    """Dynamic (3-pass) quantize — bootstrap / fallback path."""
    amax = t.abs().amax().float().clamp_(min=1e-8)
    scale = amax / _E4M3_MAX
    t8 = (t * (1.0 / scale)).to(torch.float8_e4m3fn)
    return t8, scale


class DelayedScale:
    """Per-call-site delayed scaling (transformer-engine idiom): the fused
    quant kernel writes fp8 with the PREVIOUS step's scale in ONE pass and
    accumulates this step's amax; the next scale derives from that amax
    with no separate reduction pass over the activation."""

    __slots__ = ("scale", "amax")

    def __init__(self):
        self.scale = None
        self.amax = None

    def quant(self, t):
        from ._ext import ext

        if self.scale is None:
            # bootstrap: one dynamic quantize seeds the scale
            t8, s0 = _quant(t)
            self.scale = s0.reshape(1).clone()
            self.amax = torch.zeros(1, device=t.device, dtype=torch.float32)
            return t8, self.scale.clone()
        used = self.scale.clone()  # the scale these values were written with
        t8 = ext().quant_fp8(t, self.scale, self.amax)
        # derive next step's scale from the freshly observed amax (values
        # that grew past the stale scale saturated to +-448 this step)
        torch.clamp(self.amax / _E4M3_MAX, min=1e-12, out=self.scale)
        self.amax.zero_()
        return t8, used


class _Fp8LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, x_state, w_state):
        xs = x.reshape(-1, x.shape[-1]).contiguous()
        if x_state is not None:
            x8, sx = x_state.quant(xs)
            w8, sw = w_state.quant(w)
        else:
            x8, sx = _quant(xs)
            w8, sw = _quant(w)
        y = torch._scaled_mm(x8, w8.t(), scale_a=sx, scale_b=sw,
                             bias=bias, out_dtype=x.dtype)
        ctx.save_for_backward(xs, w)
        ctx.has_bias = bias is not None
        return y.view(*x.shape[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        xs, w = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx = (dy2 @ w).view(*dy.shape[:-1], w.shape[1])
        dw = dy2.t() @ xs
        db = dy2.sum(0) if ctx.has_bias else None
        return dx, dw, db, None, None


def fp8_linear(x, w, bias=None, x_state=None, w_state=None):
    """y = x @ w^T + bias with the GEMM in fp8 e4m3 (bwd in bf16).

    Pass ``DelayedScale`` states for the single-pass fused quantize; omit
    them for the slower dynamic (3-pass) quantize."""
    return _Fp8LinearFn.apply(x, w, bias, x_state, w_state)
