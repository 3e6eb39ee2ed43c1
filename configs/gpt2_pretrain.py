"""GPT-2 pretraining on synthetic data (BASELINE config; the 345M headline
shape is hidden 1024 / 24 layers / 16 heads / seq 1024, Benchmark.md:12-26)."""

from libai_amd.config import LazyCall
from libai_amd.data import build_nlp_train_loader
from libai_amd.data.datasets import SyntheticGPTDataset

from .common.models.gpt import cfg as gpt_cfg
from .common.models.gpt import pretrain_model as model
from .common.optim import optim
from .common.train import train
from libai_amd.scheduler import WarmupCosineLR

# GPT-2 345M-class
gpt_cfg.hidden_layers = 24
gpt_cfg.hidden_size = 1024
gpt_cfg.ffn_hidden_size = 4096
gpt_cfg.num_attention_heads = 16
gpt_cfg.max_seq_length = 1024
gpt_cfg.vocab_size = 50304

dataloader = dict(
    train=LazyCall(build_nlp_train_loader)(
        dataset=LazyCall(SyntheticGPTDataset)(
            vocab_size=gpt_cfg.vocab_size,
            seq_length=gpt_cfg.max_seq_length,
            size=65536,
        ),
        train_batch_size=4,
        num_workers=2,
    ),
)

train.scheduler = LazyCall(WarmupCosineLR)(
    max_iter=1000,
    warmup_iter=100,
    warmup_factor=0.001,
    alpha=0.1,
)

train.update(
    output_dir="./output/gpt2_pretrain",
    train_micro_batch_size=4,
    train_iter=1000,
    log_period=10,
    amp=dict(enabled=True),
    dist=dict(
        data_parallel_size=None,
        tensor_parallel_size=1,
        pipeline_parallel_size=1,
        pipeline_num_layers=gpt_cfg.hidden_layers,
    ),
)
