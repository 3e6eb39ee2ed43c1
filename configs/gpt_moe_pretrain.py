"""GPT-MoE pretraining: the 345M GPT-2 trunk with expert-parallel top-2 MoE
FFNs (experts distributed over the DP/EP group, all-to-all token dispatch
over xGMI).  Beyond the reference's feature set — SURVEY §2.5 marks EP/MoE
absent upstream."""

from libai_amd.config import LazyCall
from libai_amd.scheduler import WarmupCosineLR

from .gpt2_pretrain import dataloader, gpt_cfg, model, optim, train  # noqa: F401

gpt_cfg.moe_num_experts = 8  # must be divisible by the DP(=EP) size
gpt_cfg.moe_top_k = 2

train.scheduler = LazyCall(WarmupCosineLR)(
    max_iter=1000,
    warmup_iter=100,
    warmup_factor=0.001,
    alpha=0.1,
)

train.update(
    output_dir="./output/gpt_moe_pretrain",
    # ZeRO shards over DP == the EP group; expert states are already unique
    zero_optimization=dict(enabled=False, stage=0),
)
