"""Baichuan-7B-class pretraining: Llama-family decoder (plain MHA,
rope theta 1e4, SwiGLU, RMSNorm, 64k vocab).  Reference capability:
projects/Baichuan (Baichuan on the library, Llama-shaped)."""

from libai_amd.config import LazyCall
from libai_amd.data import build_nlp_train_loader
from libai_amd.data.datasets import SyntheticGPTDataset
from libai_amd.models import LlamaForCausalLM
from libai_amd.scheduler import WarmupCosineLR

from .common.optim import optim  # noqa: F401
from .common.train import train

vocab_size = 64000
seq_len = 2048

model = LazyCall(LlamaForCausalLM)(
    hidden_layers=32,
    vocab_size=vocab_size,
    hidden_size=4096,
    intermediate_size=11008,
    num_attention_heads=32,
    num_key_value_heads=32,
    max_position_embeddings=seq_len,
    rope_theta=10000.0,
    rms_norm_eps=1e-6,
)

dataloader = dict(
    train=LazyCall(build_nlp_train_loader)(
        dataset=LazyCall(SyntheticGPTDataset)(
            vocab_size=vocab_size,
            seq_length=seq_len,
            size=65536,
        ),
        train_batch_size=2,
        num_workers=2,
    ),
)

optim.lr = 3e-4
train.scheduler = LazyCall(WarmupCosineLR)(
    max_iter=10000,
    warmup_iter=500,
    warmup_factor=0.001,
    alpha=0.1,
)

train.update(
    output_dir="./output/baichuan_7b_pretrain",
    train_micro_batch_size=2,
    train_iter=10000,
    log_period=10,
    amp=dict(enabled=True),
    activation_checkpoint=dict(enabled=True),
    zero_optimization=dict(enabled=True, stage=1),
    dist=dict(
        data_parallel_size=None,
        tensor_parallel_size=1,
        pipeline_parallel_size=1,
        pipeline_num_layers=32,
    ),
)
