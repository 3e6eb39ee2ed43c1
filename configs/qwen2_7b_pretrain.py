"""Qwen2-7B-class pretraining: Llama-family decoder with grouped-query
attention (4 KV heads shared by 28 query heads), large rope theta, tied
gated-SwiGLU MLP.  Reference capability: projects/Qwen (Qwen2 train/infer,
same shape as the Llama project)."""

from libai_amd.config import LazyCall
from libai_amd.data import build_nlp_train_loader
from libai_amd.data.datasets import SyntheticGPTDataset
from libai_amd.models import LlamaForCausalLM
from libai_amd.scheduler import WarmupCosineLR

from .common.optim import optim  # noqa: F401
from .common.train import train

vocab_size = 152064
seq_len = 2048

model = LazyCall(LlamaForCausalLM)(
    hidden_layers=28,
    vocab_size=vocab_size,
    hidden_size=3584,
    intermediate_size=18944,
    num_attention_heads=28,
    num_key_value_heads=4,
    max_position_embeddings=seq_len,
    rope_theta=1000000.0,
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
    output_dir="./output/qwen2_7b_pretrain",
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
        pipeline_num_layers=28,
    ),
)
