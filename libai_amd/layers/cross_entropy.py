"""Vocab-parallel cross-entropy loss layer (reference:
libai/layers/cross_entropy.py:26-48)."""

from torch import nn

from ..ops.cross_entropy import vocab_parallel_cross_entropy
from ..utils import distributed as du

__all__ = ["ParallelCrossEntropyLoss"]


class ParallelCrossEntropyLoss(nn.Module):
    """Per-token CE over vocab-split logits; returns losses shaped like target.

    The caller applies loss masks / reductions (the reference models do the
    same, e.g. libai/models/gpt_model.py:312-320).
    """

    def __init__(self, ignore_index=-100):
        super().__init__()
        self.ignore_index = ignore_index

    def forward(self, logits, target):
        dutil = du.get_dist_util()
        tp = dutil.tensor_parallel_size
        v_local = logits.shape[-1]
        vocab_start = dutil.tensor_parallel_rank * v_local if tp > 1 else 0
        loss = vocab_parallel_cross_entropy(
            logits, target,
            vocab_start=vocab_start,
            tp_group=dutil.tensor_parallel_group if tp > 1 else None,
            ignore_index=self.ignore_index,
        )
        return loss.view_as(target)
