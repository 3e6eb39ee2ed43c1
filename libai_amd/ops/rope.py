"""Rotary position embeddings (reference: projects/Llama/llama.py:31-64).

cos/sin tables are host-precomputed once per (max_seq, head_dim, theta) and
cached on device; the HIP kernel applies the rotation to strided
[b, s, nh, hs] views in one pass (fused into the attention prologue path).
"""

import torch

from ._ext import ext, use_hip

__all__ = ["RotaryEmbedding", "apply_rotary_pos_emb"]

_TABLE_CACHE = {}


def _tables(max_seq, head_dim, theta, device):
    key = (max_seq, head_dim, float(theta), str(device))
    if key not in _TABLE_CACHE:
        inv_freq = 1.0 / (
            theta ** (torch.arange(0, head_dim, 2, dtype=torch.float64) / head_dim)
        )
        t = torch.arange(max_seq, dtype=torch.float64)
        freqs = torch.outer(t, inv_freq)  # [s, hs/2]
        _TABLE_CACHE[key] = (
            freqs.cos().float().contiguous().to(device),
            freqs.sin().float().contiguous().to(device),
        )
    return _TABLE_CACHE[key]


class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos_t, sin_t, pos0):
        ctx.save_for_backward(cos_t, sin_t)
        ctx.pos0 = pos0
        return ext().rope(x, cos_t, sin_t, pos0, False)

    @staticmethod
    def backward(ctx, dy):
        cos_t, sin_t = ctx.saved_tensors
        return ext().rope(dy.contiguous(), cos_t, sin_t, ctx.pos0, True), None, None, None


def _ref_rope(x, cos_t, sin_t, pos0):
    s = x.shape[1]
    cos = cos_t[pos0 : pos0 + s].to(x.dtype)[None, :, None, :]
    sin = sin_t[pos0 : pos0 + s].to(x.dtype)[None, :, None, :]
    half = x.shape[-1] // 2
    x1, x2 = x[..., :half], x[..., half:]
    return torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1)


def apply_rotary_pos_emb(x, max_seq, theta=10000.0, pos0=0):
    """x: [b, s, nh, hs] (strided views ok) -> rotated contiguous tensor.

    ``pos0`` may be a DEVICE int64 tensor (hipGraph-captured decode: the
    position advances on-device, so the table rows are gathered with
    index_select instead of a host slice)."""
    cos_t, sin_t = _tables(max_seq, x.shape[-1], theta, x.device)
    if torch.is_tensor(pos0):
        s = x.shape[1]
        idx = pos0.view(1) + torch.arange(s, device=x.device)
        cos = cos_t.index_select(0, idx)[None, :, None, :]
        sin = sin_t.index_select(0, idx)[None, :, None, :]
        half = x.shape[-1] // 2
        # fp32 math + one final cast: matches the HIP kernel's rounding
        x1, x2 = x[..., :half].float(), x[..., half:].float()
        return torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin],
                         dim=-1).to(x.dtype)
    if use_hip(x):
        return _RopeFn.apply(x, cos_t, sin_t, pos0)
    return _ref_rope(x, cos_t, sin_t, pos0)


class RotaryEmbedding(torch.nn.Module):
    def __init__(self, head_dim, max_seq_length, theta=10000.0):
        super().__init__()
        self.head_dim = head_dim
        self.max_seq = max_seq_length
        self.theta = theta

    def forward(self, x, pos0=0):
        return apply_rotary_pos_emb(x, self.max_seq, self.theta, pos0)
