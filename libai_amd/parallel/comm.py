"""Autograd-aware collectives for tensor parallelism.

These materialize the implicit SBP re-signings of the reference as explicit
RCCL calls (reference: libai/layers/linear.py:122-168
grad_sbp annotations):

  * copy_to_tensor_parallel_region    — forward identity, backward all-reduce
    (the input side of a column-parallel linear, C2)
  * reduce_from_tensor_parallel_region — forward all-reduce, backward identity
    (the output side of a row-parallel linear, C1)
  * gather_from_tensor_parallel_region / scatter_to_tensor_parallel_region —
    last-dim all-gather / split pairs.
"""

import torch
import torch.distributed as dist

from ..utils import distributed as du

__all__ = [
    "copy_to_tensor_parallel_region",
    "reduce_from_tensor_parallel_region",
    "gather_from_tensor_parallel_region",
    "scatter_to_tensor_parallel_region",
    "all_reduce_sum_differentiable",
    "gather_from_sequence_parallel_region",
    "reduce_scatter_to_sequence_parallel_region",
    "scatter_to_sequence_parallel_region",
]


def _tp_group():
    return du.get_dist_util().tensor_parallel_group


def _tp_size():
    return du.get_dist_util().tensor_parallel_size


def _all_reduce(x, group):
    if dist.is_initialized():
        x = x.contiguous()
        dist.all_reduce(x, group=group)
    return x


class _CopyToTP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        return x

    @staticmethod
    def backward(ctx, grad):
        return _all_reduce(grad.clone(), _tp_group())


class _ReduceFromTP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        return _all_reduce(x, _tp_group())

    @staticmethod
    def backward(ctx, grad):
        return grad


class _GatherFromTP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        tp = _tp_size()
        if tp == 1:
            return x
        x = x.contiguous()
        parts = [torch.empty_like(x) for _ in range(tp)]
        dist.all_gather(parts, x, group=_tp_group())
        return torch.cat(parts, dim=-1)

    @staticmethod
    def backward(ctx, grad):
        tp = _tp_size()
        if tp == 1:
            return grad
        rank = du.get_dist_util().tensor_parallel_rank
        return grad.chunk(tp, dim=-1)[rank].contiguous()


class _ScatterToTP(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        tp = _tp_size()
        if tp == 1:
            return x
        rank = du.get_dist_util().tensor_parallel_rank
        return x.chunk(tp, dim=-1)[rank].contiguous()

    @staticmethod
    def backward(ctx, grad):
        tp = _tp_size()
        if tp == 1:
            return grad
        grad = grad.contiguous()
        parts = [torch.empty_like(grad) for _ in range(tp)]
        dist.all_gather(parts, grad, group=_tp_group())
        return torch.cat(parts, dim=-1)


class _AllReduceSumDiff(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        return _all_reduce(x.clone(), group)

    @staticmethod
    def backward(ctx, grad):
        return grad, None


# -- sequence parallelism (Megatron-SP; beyond the reference's feature set:
# SURVEY §2.5 marks SP absent upstream).  Activations in the LN/dropout
# regions are sharded along the SEQUENCE dim (dim 1 of [b, s, h]); entering
# a TP block they are all-gathered, and the TP block's partial-sum output is
# reduce-scattered back to shards — same bytes on the wire as the C1/C2
# all-reduces, but LN/dropout compute and activation memory divide by tp.


def _seq_chunks(x, tp):
    return x.chunk(tp, dim=1)


class _GatherSeq(torch.autograd.Function):
    """fwd: all-gather along seq; bwd: reduce-scatter (sums the per-rank
    partial full-sequence grads, returns this rank's shard)."""

    @staticmethod
    def forward(ctx, x):
        tp = _tp_size()
        x = x.contiguous()
        parts = [torch.empty_like(x) for _ in range(tp)]
        dist.all_gather(parts, x, group=_tp_group())
        return torch.cat(parts, dim=1)

    @staticmethod
    def backward(ctx, grad):
        tp = _tp_size()
        grad = grad.contiguous()
        chunks = list(_seq_chunks(grad, tp))
        out = torch.empty_like(chunks[0])
        dist.reduce_scatter(out, [c.contiguous() for c in chunks],
                            group=_tp_group())
        return out


class _ReduceScatterSeq(torch.autograd.Function):
    """fwd: reduce-scatter along seq (the row-linear partial sums); bwd:
    all-gather."""

    @staticmethod
    def forward(ctx, x):
        tp = _tp_size()
        x = x.contiguous()
        chunks = [c.contiguous() for c in _seq_chunks(x, tp)]
        out = torch.empty_like(chunks[0])
        dist.reduce_scatter(out, chunks, group=_tp_group())
        return out

    @staticmethod
    def backward(ctx, grad):
        tp = _tp_size()
        grad = grad.contiguous()
        parts = [torch.empty_like(grad) for _ in range(tp)]
        dist.all_gather(parts, grad, group=_tp_group())
        return torch.cat(parts, dim=1)


class _ScatterSeq(torch.autograd.Function):
    """fwd: keep this rank's seq shard (replicated input); bwd: all-gather."""

    @staticmethod
    def forward(ctx, x):
        tp = _tp_size()
        rank = du.get_dist_util().tensor_parallel_rank
        return _seq_chunks(x, tp)[rank].contiguous()

    @staticmethod
    def backward(ctx, grad):
        tp = _tp_size()
        grad = grad.contiguous()
        parts = [torch.empty_like(grad) for _ in range(tp)]
        dist.all_gather(parts, grad, group=_tp_group())
        return torch.cat(parts, dim=1)


def gather_from_sequence_parallel_region(x):
    if _tp_size() == 1:
        return x
    return _GatherSeq.apply(x)


def reduce_scatter_to_sequence_parallel_region(x):
    if _tp_size() == 1:
        return x
    return _ReduceScatterSeq.apply(x)


def scatter_to_sequence_parallel_region(x):
    if _tp_size() == 1:
        return x
    return _ScatterSeq.apply(x)


def copy_to_tensor_parallel_region(x):
    if _tp_size() == 1:
        return x
    return _CopyToTP.apply(x)


def reduce_from_tensor_parallel_region(x):
    if _tp_size() == 1:
        return x
    return _ReduceFromTP.apply(x)


def gather_from_tensor_parallel_region(x):
    return _GatherFromTP.apply(x)


def scatter_to_tensor_parallel_region(x):
    return _ScatterToTP.apply(x)


def all_reduce_sum_differentiable(x, group=None):
    if not dist.is_initialized():
        return x
    return _AllReduceSumDiff.apply(x, group if group is not None else _tp_group())
