"""Vocab-parallel cross entropy.

Replaces the reference's flow._C.sparse_softmax_cross_entropy on vocab-split
logits (reference: libai/layers/cross_entropy.py:26-48).  The softmax is never
materialized; TP reduction is two allreduces over [R]-sized stat tensors
(max, then corrected sum-exp + target logit) — C4 in SURVEY.md §2.4.
"""

import torch
import torch.distributed as dist

from ._ext import ext, use_hip

__all__ = ["vocab_parallel_cross_entropy"]


class _VocabParallelCEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target, vocab_start, tp_group, ignore_index):
        # logits: [R, V_local]; target: [R] global vocab ids
        logits = logits.contiguous()
        target = target.contiguous()
        lmax, lsumexp, tlogit = ext().ce_fwd(logits, target, vocab_start, ignore_index)
        if tp_group is not None and dist.is_initialized():
            gmax = lmax.clone()
            dist.all_reduce(gmax, op=dist.ReduceOp.MAX, group=tp_group)
            stats = torch.stack([lsumexp * torch.exp(lmax - gmax), tlogit])
            dist.all_reduce(stats, group=tp_group)
            gsumexp, gtlogit = stats[0], stats[1]
        else:
            gmax, gsumexp, gtlogit = lmax, lsumexp, tlogit
        loss = torch.log(gsumexp) + gmax - gtlogit
        ignored = target == ignore_index
        loss = loss.masked_fill(ignored, 0.0)
        ctx.save_for_backward(logits, target, gmax, gsumexp)
        ctx.vocab_start = vocab_start
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, target, gmax, gsumexp = ctx.saved_tensors
        dlogits = ext().ce_bwd(
            logits, target, gmax, gsumexp, dloss.contiguous().float(),
            ctx.vocab_start, ctx.ignore_index,
        )
        return dlogits, None, None, None, None


def _ref(logits, target, vocab_start, tp_group, ignore_index):
    from ..parallel.comm import all_reduce_sum_differentiable

    lf = logits.float()
    lmax = lf.max(dim=-1).values.detach()  # softmax is max-shift invariant
    if tp_group is not None and dist.is_initialized():
        gmax = lmax.clone()
        dist.all_reduce(gmax, op=dist.ReduceOp.MAX, group=tp_group)
    else:
        gmax = lmax
    sumexp = torch.exp(lf - gmax[:, None]).sum(-1)
    local_t = target - vocab_start
    in_shard = (local_t >= 0) & (local_t < logits.shape[-1]) & (target != ignore_index)
    tlogit = torch.where(
        in_shard, lf.gather(1, local_t.clamp(0, logits.shape[-1] - 1)[:, None])[:, 0],
        torch.zeros_like(gmax),
    )
    if tp_group is not None and dist.is_initialized():
        stats = torch.stack([sumexp, tlogit])
        stats = all_reduce_sum_differentiable(stats, tp_group)
        sumexp, tlogit = stats[0], stats[1]
    loss = torch.log(sumexp) + gmax - tlogit
    return loss * (target != ignore_index).to(loss.dtype)


def vocab_parallel_cross_entropy(logits, target, vocab_start=0, tp_group=None,
                                 ignore_index=-100):
    """Per-token CE loss [R] from vocab-sharded logits [R, V_local].

    target holds GLOBAL vocab ids; each rank contributes its shard.
    """
    R = target.numel()
    logits2d = logits.reshape(R, logits.shape[-1])
    target1d = target.reshape(R)
    if use_hip(logits):
        return _VocabParallelCEFn.apply(logits2d, target1d, vocab_start, tp_group,
                                        ignore_index)
    return _ref(logits2d, target1d, vocab_start, tp_group, ignore_index)
