"""LoRA adapters (reference capability: projects/ChatGLM/lora/layers.py:198-237
and lora_model.py — the reference ships LoRA as a downstream project; here it
is a first-class utility that wraps libai_amd layers).

``apply_lora(model, r, alpha, target_modules)`` wraps matching Linear1D /
Conv1D modules with trainable low-rank A/B factors and freezes everything
else; ``merge_lora`` folds the update back into the base weight.

TP note: for a column-parallel base the B factor is sharded on its output
dim (A replicated); for a row-parallel base the A factor is sharded on its
input dim (B replicated) — matching the base layer's shard geometry.  The
rank-r bottleneck needs one extra TP collective on the [.., r] activation:
all-reduce forward for the row case (partial input-dim sum) and all-reduce
backward for the col case (partial grads toward replicated A).
"""

import math
import re

import torch
from torch import nn

from .layers.linear import Linear1D
from .parallel.comm import (
    copy_to_tensor_parallel_region,
    reduce_from_tensor_parallel_region,
)
from .utils import distributed as du

__all__ = ["LoRALinear", "apply_lora", "merge_lora", "mark_only_lora_as_trainable"]


class LoRALinear(nn.Module):
    def __init__(self, base: Linear1D, r=8, alpha=16, dropout=0.0):
        super().__init__()
        self.base = base
        self.r = r
        self.scaling = alpha / r
        self.lora_dropout = nn.Dropout(dropout)
        dutil = du.get_dist_util()
        tp = dutil.tensor_parallel_size
        in_f, out_f = base.in_features, base.out_features
        dtype = base.weight.dtype
        device = base.weight.device

        if base.parallel == "col":
            a_in, b_out = in_f, out_f // tp      # B sharded like the base output
        elif base.parallel == "row":
            a_in, b_out = in_f // tp, out_f      # A sharded like the base input
        else:
            a_in, b_out = in_f, out_f
        self.lora_A = nn.Parameter(torch.zeros(r, a_in, dtype=dtype, device=device))
        self.lora_B = nn.Parameter(torch.zeros(b_out, r, dtype=dtype, device=device))
        if base.parallel == "col":
            self.lora_B.tensor_parallel = True
            self.lora_B.tp_shard_dim = 0
            # inherit fused-projection pairing (e.g. [gate|up]) so checkpoint
            # consolidation/reshard keeps the canonical layout
            self.lora_B.tp_fused_chunks = getattr(base.weight, "tp_fused_chunks", 1)
        elif base.parallel == "row":
            self.lora_A.tensor_parallel = True
            self.lora_A.tp_shard_dim = 1
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
        self.base.weight.requires_grad_(False)
        if self.base.bias is not None:
            self.base.bias.requires_grad_(False)

    def forward(self, x):
        out = self.base(x)
        h = self.lora_dropout(x) @ self.lora_A.t()
        if self.base.parallel == "row":
            # A is sharded on the input dim, so h is a PARTIAL sum over TP:
            # all-reduce in forward (identity backward) before the replicated
            # B projection — mirrors the base row-linear's C1 collective.
            h = reduce_from_tensor_parallel_region(h)
        elif self.base.parallel == "col":
            # h is replicated but the grad arriving from the sharded-B branch
            # is partial per rank; identity-fwd/all-reduce-bwd here keeps the
            # replicated lora_A (and x) gradients coherent across TP ranks.
            h = copy_to_tensor_parallel_region(h)
        delta = h @ self.lora_B.t() * self.scaling
        if isinstance(out, tuple):  # skip_bias_add base
            return out[0] + delta, out[1]
        return out + delta

    @torch.no_grad()
    def merge(self):
        self.base.weight += (self.lora_B @ self.lora_A) * self.scaling
        return self.base


def apply_lora(model, r=8, alpha=16, dropout=0.0,
               target_modules=(r"query_key_value", r"dense", r"o_proj",
                               r"gate_up_proj", r"down_proj")):
    """Wrap matching Linear1D submodules in-place; returns the model."""
    patterns = [re.compile(p) for p in target_modules]
    replaced = 0
    for name, module in list(model.named_modules()):
        for child_name, child in list(module.named_children()):
            if isinstance(child, Linear1D) and any(
                p.search(child_name) for p in patterns
            ):
                setattr(module, child_name, LoRALinear(child, r, alpha, dropout))
                replaced += 1
    if replaced == 0:
        raise ValueError(f"no modules matched {target_modules}")
    mark_only_lora_as_trainable(model)
    return model


def mark_only_lora_as_trainable(model, bias="none"):
    for n, p in model.named_parameters():
        p.requires_grad_("lora_" in n or (bias == "all" and n.endswith("bias")))


def merge_lora(model):
    """Fold every LoRALinear back into its base layer (for inference)."""
    for name, module in list(model.named_modules()):
        for child_name, child in list(module.named_children()):
            if isinstance(child, LoRALinear):
                setattr(module, child_name, child.merge())
    return model
