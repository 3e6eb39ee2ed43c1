"""GPT-2 345M 3D-parallel (BASELINE config #3: TP=2 PP=2 DP=2 on 8 GPUs)."""

from .gpt2_pretrain import dataloader, model, optim, train
from .gpt2_pretrain import gpt_cfg

train.update(
    output_dir="./output/gpt2_3d_pretrain",
    train_micro_batch_size=4,
    num_accumulation_steps=8,  # = 1F1B micro-batch count
    dist=dict(
        data_parallel_size=2,
        tensor_parallel_size=2,
        pipeline_parallel_size=2,
        pipeline_num_layers=gpt_cfg.hidden_layers,
    ),
)

# Megatron-style sequence parallelism (beyond the reference's feature set):
# shard the LN/dropout regions along seq over the TP group —
#   model.cfg.sequence_parallel = True   (requires pipeline_parallel_size=1)
