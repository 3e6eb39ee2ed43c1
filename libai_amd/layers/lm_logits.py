"""Tied-embedding output projection (reference: libai/layers/lm_logits.py:44-58).

logits = hidden @ embedding_weight^T with the embedding weight vocab-sharded:
each rank produces a vocab-split logits slice [.., v/tp] that feeds the
vocab-parallel cross entropy.  The input copy into the TP region pins the
backward all-reduce the reference encoded as grad_sbp (lm_logits.py:56, C2).
"""

import torch.nn.functional as F
from torch import nn

from ..parallel.comm import copy_to_tensor_parallel_region
from ..utils import distributed as du
from .linear import Linear1D

__all__ = ["LMLogits"]


class LMLogits(nn.Module):
    def __init__(self, vocab_size, bias=False, sequence_parallel=False, *,
                 layer_idx=-1):
        super().__init__()
        self.vocab_size = vocab_size
        self.layer_idx = layer_idx
        self.sequence_parallel = sequence_parallel
        if bias:
            dutil = du.get_dist_util()
            tp = dutil.tensor_parallel_size
            import torch

            self.bias = nn.Parameter(torch.zeros(vocab_size // tp))
            self.bias.tensor_parallel = True
            self.bias.tp_shard_dim = 0
        else:
            self.register_parameter("bias", None)

    def forward(self, hidden_states, word_embeddings_weight):
        if self.sequence_parallel:
            # SP: hidden arrives seq-sharded; the all-gather's backward
            # reduce-scatter replaces the copy_to all-reduce
            from ..parallel.comm import gather_from_sequence_parallel_region

            x = gather_from_sequence_parallel_region(hidden_states)
        else:
            x = copy_to_tensor_parallel_region(hidden_states)
        logits = F.linear(x, word_embeddings_weight, self.bias)
        return logits
