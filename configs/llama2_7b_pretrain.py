"""Llama-2 7B pretraining (BASELINE config #4: TP4 PP2 + ZeRO-2 + act ckpt)."""

from libai_amd.config import LazyCall
from libai_amd.data import build_nlp_train_loader
from libai_amd.data.datasets import SyntheticGPTDataset

from .common.models.llama import cfg as llama_cfg
from .common.models.llama import model
from .common.optim import optim
from .common.train import train

dataloader = dict(
    train=LazyCall(build_nlp_train_loader)(
        dataset=LazyCall(SyntheticGPTDataset)(
            vocab_size=llama_cfg.vocab_size,
            seq_length=llama_cfg.max_position_embeddings,
            size=65536,
        ),
        train_batch_size=1,
        num_workers=2,
    ),
)

train.update(
    output_dir="./output/llama2_7b_pretrain",
    train_micro_batch_size=1,
    num_accumulation_steps=8,
    train_iter=1000,
    log_period=5,
    amp=dict(enabled=True),
    activation_checkpoint=dict(enabled=True),
    zero_optimization=dict(enabled=True, stage=2),
    dist=dict(
        data_parallel_size=1,
        tensor_parallel_size=4,
        pipeline_parallel_size=2,
        pipeline_num_layers=llama_cfg.hidden_layers,
    ),
)
