"""TP-shard a HuggingFace transformers model for inference.

The reference's mock_transformers project ran HF models on OneFlow via
flow.mock_torch and hand-edited sbp per module
(projects/mock_transformers/dist_infer_llama.py:93-124).  On PyTorch-ROCm HF
models already run natively, so the equivalent capability is this utility:
walk an HF module, shard the attention/MLP linears column/row-wise across
the TP group, and insert the explicit collectives.

Usage (same model code on every rank, launched one rank per GPU):

    from transformers import AutoModelForCausalLM
    model = AutoModelForCausalLM.from_pretrained(...)
    tp_shard_hf_model(model)   # in-place; row outputs all-reduce over TP
"""

import logging
import re

import torch
from torch import nn

from ...parallel.comm import (
    copy_to_tensor_parallel_region,
    reduce_from_tensor_parallel_region,
)
from ...utils import distributed as du

__all__ = ["tp_shard_hf_model", "COL_PATTERNS", "ROW_PATTERNS"]

logger = logging.getLogger(__name__)

# output-sharded (column-parallel) projections across common HF architectures
COL_PATTERNS = (
    r"q_proj", r"k_proj", r"v_proj", r"gate_proj", r"up_proj",  # llama/qwen/mistral
    r"query", r"key", r"value", r"intermediate\.dense",         # bert
    r"c_fc",                                                    # gpt2 mlp in
    r"fc1", r"wi_0", r"wi_1", r"wi",                            # t5/opt
)
# input-sharded (row-parallel) projections whose outputs need an all-reduce
ROW_PATTERNS = (
    r"o_proj", r"down_proj",
    r"attention\.output\.dense", r"output\.dense",
    r"c_proj",
    r"fc2", r"wo",
)


class _ColShardLinear(nn.Module):
    def __init__(self, base: nn.Linear, tp, tpr):
        super().__init__()
        assert base.out_features % tp == 0, (base.out_features, tp)
        w = base.weight.data.chunk(tp, dim=0)[tpr].clone()
        self.weight = nn.Parameter(w, requires_grad=False)
        if base.bias is not None:
            self.bias = nn.Parameter(base.bias.data.chunk(tp, 0)[tpr].clone(),
                                     requires_grad=False)
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        x = copy_to_tensor_parallel_region(x)
        return torch.nn.functional.linear(x, self.weight, self.bias)


class _RowShardLinear(nn.Module):
    def __init__(self, base: nn.Linear, tp, tpr):
        super().__init__()
        assert base.in_features % tp == 0, (base.in_features, tp)
        w = base.weight.data.chunk(tp, dim=1)[tpr].clone()
        self.weight = nn.Parameter(w, requires_grad=False)
        if base.bias is not None:  # added once, after the reduction
            self.bias = nn.Parameter(base.bias.data.clone(), requires_grad=False)
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        out = torch.nn.functional.linear(x, self.weight)
        out = reduce_from_tensor_parallel_region(out)
        if self.bias is not None:
            out = out + self.bias
        return out


def tp_shard_hf_model(model, col_patterns=COL_PATTERNS, row_patterns=ROW_PATTERNS,
                      fix_num_heads_attrs=("num_heads", "num_attention_heads",
                                           "num_key_value_heads")):
    """Shard matching nn.Linear submodules in place across the TP group.

    Also divides any per-module head-count attributes by tp so HF attention
    reshape logic keeps working on the sharded hidden dim.
    """
    dutil = du.get_dist_util()
    tp, tpr = dutil.tensor_parallel_size, dutil.tensor_parallel_rank
    if tp == 1:
        return model
    col = [re.compile(p) for p in col_patterns]
    row = [re.compile(p) for p in row_patterns]
    n_col = n_row = 0
    for name, module in list(model.named_modules()):
        for child_name, child in list(module.named_children()):
            if not isinstance(child, nn.Linear):
                continue
            full = f"{name}.{child_name}" if name else child_name
            if any(p.search(full) for p in row):
                setattr(module, child_name, _RowShardLinear(child, tp, tpr))
                n_row += 1
            elif any(p.search(full) for p in col):
                setattr(module, child_name, _ColShardLinear(child, tp, tpr))
                n_col += 1
    # fix head counts for reshape logic inside HF attention modules
    for m in model.modules():
        for attr in fix_num_heads_attrs:
            v = getattr(m, attr, None)
            if isinstance(v, int) and v % tp == 0 and v > 1:
                setattr(m, attr, v // tp)
    if n_col == 0 and n_row == 0:
        raise ValueError("no linear layers matched the TP shard patterns")
    logger.info(f"TP-sharded HF model: {n_col} column + {n_row} row linears, tp={tp}")
    return model
