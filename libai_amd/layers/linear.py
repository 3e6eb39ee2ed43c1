"""Column/row/data-parallel linear layers.

Explicit-collective re-design of the reference's SBP-annotated Linear1D
(reference: libai/layers/linear.py:78-168):

  * ``parallel="col"``: weight sharded along the OUTPUT dim ([out/tp, in]).
    Forward copies the (replicated) input into the TP region (identity fwd /
    all-reduce bwd — the reference's grad_sbp pin at linear.py:131, C2 in
    SURVEY.md §2.4) and produces a split output.
  * ``parallel="row"``: weight sharded along the INPUT dim ([out, in/tp]).
    The partial output is summed with an all-reduce (fwd) / identity (bwd)
    (C1); the bias is added AFTER the reduction.
  * ``parallel="data"``: plain replicated linear.

``skip_bias_add=True`` returns (output, bias) so the caller can fuse the bias
into a following elementwise kernel (bias_gelu / bias_dropout_add).
"""

import torch
import torch.nn.functional as F
from torch import nn

from ..parallel.comm import (
    copy_to_tensor_parallel_region,
    reduce_from_tensor_parallel_region,
)
from ..utils import distributed as du

__all__ = ["Linear1D", "Linear"]


def tp_slice(full, tp, tpr, dim, fused_chunks=1):
    """This rank's TP shard of a full tensor.

    ``fused_chunks > 1`` handles fused projections like [gate | up]: the
    canonical full layout is the concatenation of the chunks, but each rank
    must hold a PAIRED slice ([gate_r | up_r]) -- a plain contiguous chunk
    would give rank 0 all of gate and rank 1 all of up, silently changing
    the function the gated MLP computes under TP.
    """
    if fused_chunks <= 1:
        return full.chunk(tp, dim=dim)[tpr]
    parts = full.chunk(fused_chunks, dim=dim)
    return torch.cat([h.chunk(tp, dim=dim)[tpr] for h in parts], dim=dim)


def tp_merge(shards, dim, fused_chunks=1):
    """Inverse of tp_slice: per-rank shards -> canonical full tensor."""
    if fused_chunks <= 1:
        return torch.cat(shards, dim=dim)
    per = [s.chunk(fused_chunks, dim=dim) for s in shards]
    return torch.cat(
        [torch.cat([p[h] for p in per], dim=dim) for h in range(fused_chunks)],
        dim=dim,
    )


def init_tp_shard_(param, full_shape, init_method, shard_dim, fused_chunks=1):
    """Initialize a TP-sharded parameter as a SLICE of the full-tensor init.

    All TP ranks draw the same full tensor (they run init under the same seed)
    and keep their shard, so sharded == replicated numerics hold exactly —
    the reference gets this from global-tensor init (libai/layers/linear.py:98-105).
    """
    dutil = du.get_dist_util()
    tp, tpr = dutil.tensor_parallel_size, dutil.tensor_parallel_rank
    full = torch.empty(*full_shape, dtype=param.dtype, device=param.device)
    init_method(full)
    if tp == 1:
        with torch.no_grad():
            param.copy_(full)
        return
    shard = tp_slice(full, tp, tpr, shard_dim, fused_chunks)
    with torch.no_grad():
        param.copy_(shard)


class Linear1D(nn.Module):
    def __init__(
        self,
        in_features,
        out_features,
        bias=True,
        parallel="data",
        init_method=nn.init.xavier_normal_,
        skip_bias_add=False,
        fused_chunks=1,
        sequence_parallel=False,
        *,
        layer_idx=0,
        dtype=None,
    ):
        super().__init__()
        assert parallel in ("data", "col", "row"), parallel
        self.in_features = in_features
        self.out_features = out_features
        self.parallel = parallel
        self.skip_bias_add = skip_bias_add
        self.sequence_parallel = sequence_parallel
        self.layer_idx = layer_idx

        dutil = du.get_dist_util()
        tp = dutil.tensor_parallel_size
        dtype = dtype or torch.get_default_dtype()

        if parallel == "col":
            assert out_features % tp == 0, (out_features, tp)
            self.weight = nn.Parameter(
                torch.empty(out_features // tp, in_features, dtype=dtype)
            )
            self.weight.tensor_parallel = True
            self.weight.tp_shard_dim = 0
            self.weight.tp_fused_chunks = fused_chunks
            init_tp_shard_(self.weight, (out_features, in_features), init_method, 0,
                           fused_chunks)
            if bias:
                self.bias = nn.Parameter(torch.zeros(out_features // tp, dtype=dtype))
                self.bias.tensor_parallel = True
                self.bias.tp_shard_dim = 0
                self.bias.tp_fused_chunks = fused_chunks
            else:
                self.register_parameter("bias", None)
        elif parallel == "row":
            assert in_features % tp == 0, (in_features, tp)
            self.weight = nn.Parameter(
                torch.empty(out_features, in_features // tp, dtype=dtype)
            )
            self.weight.tensor_parallel = True
            self.weight.tp_shard_dim = 1
            init_tp_shard_(self.weight, (out_features, in_features), init_method, 1)
            if bias:
                self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype))
                if sequence_parallel:
                    # bias applies to seq-SHARDED outputs: its grad is a
                    # partial sum over this rank's tokens -> TP all-reduce
                    # at grad sync (FusedAdamW._sync_sp_grads)
                    self.bias.sequence_parallel_grad = True
            else:
                self.register_parameter("bias", None)
        else:
            self.weight = nn.Parameter(torch.empty(out_features, in_features, dtype=dtype))
            init_method(self.weight.data)
            if bias:
                self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype))
            else:
                self.register_parameter("bias", None)

    def _gemm(self, x, bias=None):
        from ..ops import fp8

        if fp8.fp8_eligible(x, self.weight):
            if not hasattr(self, "_fp8_states"):
                self._fp8_states = (fp8.DelayedScale(), fp8.DelayedScale())
            xs, ws = self._fp8_states
            return fp8.fp8_linear(x, self.weight, bias, x_state=xs,
                                  w_state=ws)
        return F.linear(x, self.weight, bias)

    def forward(self, x):
        if self.parallel == "col":
            if self.sequence_parallel:
                # SP: input arrives seq-sharded; all-gather replaces the
                # copy_to (its backward reduce-scatters the partial grads)
                from ..parallel.comm import gather_from_sequence_parallel_region

                x = gather_from_sequence_parallel_region(x)
            else:
                x = copy_to_tensor_parallel_region(x)
            if self.skip_bias_add:
                return self._gemm(x), self.bias
            # bias fused into the GEMM epilogue (a separate add costs a full
            # HBM round-trip over the 3h-wide qkv activations)
            return self._gemm(x, self.bias)
        if self.parallel == "row":
            out = self._gemm(x)
            if self.sequence_parallel:
                # SP: reduce the TP partial sums AND scatter back to shards
                from ..parallel.comm import (
                    reduce_scatter_to_sequence_parallel_region,
                )

                out = reduce_scatter_to_sequence_parallel_region(out)
            else:
                out = reduce_from_tensor_parallel_region(out)
            if self.skip_bias_add:
                return out, self.bias
            if self.bias is not None:
                out = out + self.bias
            return out
        if self.skip_bias_add:
            return self._gemm(x), self.bias
        return self._gemm(x, self.bias)

    def extra_repr(self):
        return (
            f"in_features={self.in_features}, out_features={self.out_features}, "
            f"parallel={self.parallel}, layer_idx={self.layer_idx}"
        )


Linear = Linear1D
