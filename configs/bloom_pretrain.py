"""BLOOM pretraining recipe (ALiBi attention, embedding LayerNorm, tied
logits) on synthetic data.  Reference capability: projects/BLOOM."""

from libai_amd.config import LazyCall
from libai_amd.data import build_nlp_train_loader
from libai_amd.data.datasets import SyntheticGPTDataset
from libai_amd.models import BloomForCausalLM
from libai_amd.scheduler import WarmupCosineLR

from .common.optim import optim  # noqa: F401
from .common.train import train

vocab_size = 50304
seq_len = 1024

model = LazyCall(BloomForCausalLM)(
    vocab_size=vocab_size,
    hidden_size=1024,
    hidden_layers=24,
    num_attention_heads=16,
    hidden_dropout_prob=0.1,
    attention_dropout_prob=0.1,
)

dataloader = dict(
    train=LazyCall(build_nlp_train_loader)(
        dataset=LazyCall(SyntheticGPTDataset)(
            vocab_size=vocab_size,
            seq_length=seq_len,
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
    output_dir="./output/bloom_pretrain",
    train_micro_batch_size=4,
    train_iter=1000,
    log_period=10,
    amp=dict(enabled=True),
    dist=dict(
        data_parallel_size=None,
        tensor_parallel_size=1,
        pipeline_parallel_size=1,
        pipeline_num_layers=24,
    ),
)
