"""Additive attention score biases: T5 relative-position bias and ALiBi.

Reference capabilities: the MT5 project's relative position bias
(reference projects/MT5/layers/attention_layer.py:118-123) and BLOOM's
ALiBi attention (reference projects/BLOOM/modeling/).  Both feed the
``position_bias`` argument of MultiheadAttention / TransformerLayer.
"""

import math

import torch
from torch import nn

from ..utils import distributed as du

__all__ = ["T5RelativePositionBias", "build_alibi_bias", "alibi_slopes"]


class T5RelativePositionBias(nn.Module):
    """Learned bucketed relative-position bias [1, nh_local, sq, sk].

    The [num_buckets, num_heads] table is sharded over the TP head dim like
    the attention heads themselves."""

    def __init__(self, num_heads, num_buckets=32, max_distance=128,
                 bidirectional=True, init_method=nn.init.xavier_normal_,
                 *, layer_idx=0):
        super().__init__()
        dutil = du.get_dist_util()
        tp = dutil.tensor_parallel_size
        assert num_heads % tp == 0
        self.num_buckets = num_buckets
        self.max_distance = max_distance
        self.bidirectional = bidirectional
        self.layer_idx = layer_idx
        self.weight = nn.Parameter(
            torch.empty(num_buckets, num_heads // tp)
        )
        self.weight.tensor_parallel = True
        self.weight.tp_shard_dim = 1
        from .linear import init_tp_shard_

        init_tp_shard_(self.weight, (num_buckets, num_heads), init_method, 1)

    @staticmethod
    def _bucket(rel_pos, bidirectional, num_buckets, max_distance):
        # T5 bucketing: half the buckets exact, half log-spaced to max_distance
        ret = torch.zeros_like(rel_pos)
        n = -rel_pos
        if bidirectional:
            num_buckets //= 2
            ret = ret + (n < 0).long() * num_buckets
            n = n.abs()
        else:
            n = torch.clamp(n, min=0)
        max_exact = num_buckets // 2
        is_small = n < max_exact
        large = max_exact + (
            torch.log(n.float().clamp(min=1) / max_exact)
            / math.log(max_distance / max_exact)
            * (num_buckets - max_exact)
        ).long()
        large = torch.clamp(large, max=num_buckets - 1)
        return ret + torch.where(is_small, n, large)

    def forward(self, sq, sk, device=None):
        device = device or self.weight.device
        ctx = torch.arange(sq, device=device)[:, None]
        mem = torch.arange(sk, device=device)[None, :]
        buckets = self._bucket(mem - ctx, self.bidirectional, self.num_buckets,
                               self.max_distance)  # [sq, sk]
        vals = self.weight[buckets]  # [sq, sk, nh_local]
        return vals.permute(2, 0, 1).unsqueeze(0).to(self.weight.dtype)


def alibi_slopes(num_heads):
    """BLOOM's per-head geometric slope schedule (closest power of 2)."""
    closest = 2 ** math.floor(math.log2(num_heads))
    base = 2.0 ** (-(2.0 ** -(math.log2(closest) - 3)))
    slopes = [base ** (i + 1) for i in range(closest)]
    if closest != num_heads:
        extra_base = 2.0 ** (-(2.0 ** -(math.log2(2 * closest) - 3)))
        slopes += [extra_base ** (i + 1) for i in range(0, 2 * (num_heads - closest), 2)]
    return torch.tensor(slopes)


def build_alibi_bias(num_heads, seq_len, device=None, dtype=torch.float32):
    """[nh_local, 1, seq] ALiBi bias (this TP rank's head slice).

    slope_h * j is row-shift-equivalent to slope_h * (j - i) under softmax,
    so the per-query term is dropped (BLOOM's own formulation)."""
    dutil = du.get_dist_util()
    tp, tpr = dutil.tensor_parallel_size, dutil.tensor_parallel_rank
    slopes = alibi_slopes(num_heads).to(device=device, dtype=torch.float32)
    local = slopes.chunk(tp)[tpr] if tp > 1 else slopes
    pos = torch.arange(seq_len, device=device, dtype=torch.float32)
    return (local[:, None, None] * pos[None, None, :]).to(dtype)
