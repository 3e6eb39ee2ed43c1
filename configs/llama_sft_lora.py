"""Llama supervised fine-tuning with LoRA adapters (reference capability:
projects/Llama SFT + projects/ChatGLM lora)."""

from libai_amd.config import LazyCall
from libai_amd.data import build_nlp_train_loader
from libai_amd.data.datasets import SyntheticGPTDataset

from .common.models.llama import cfg as llama_cfg
from .common.models.llama import model
from .common.optim import optim
from .common.train import train

# smaller demo shape; point model.cfg at 7B + train.load_weight at a
# pretrained checkpoint for real SFT
llama_cfg.hidden_layers = 8
llama_cfg.hidden_size = 1024
llama_cfg.intermediate_size = 2816
llama_cfg.num_attention_heads = 16

dataloader = dict(
    train=LazyCall(build_nlp_train_loader)(
        dataset=LazyCall(SyntheticGPTDataset)(
            vocab_size=llama_cfg.vocab_size,
            seq_length=512,
            size=8192,
        ),
        train_batch_size=4,
        num_workers=2,
    ),
)

optim.lr = 1e-4
optim.params.clip_grad_max_norm = 1.0

train.update(
    output_dir="./output/llama_sft_lora",
    train_micro_batch_size=4,
    train_iter=500,
    amp=dict(enabled=True),
    # LoRA: rank/alpha consumed by tools/train_net.py via train.lora
    lora=dict(enabled=True, r=16, alpha=32, dropout=0.05),
    dist=dict(
        data_parallel_size=None,
        tensor_parallel_size=1,
        pipeline_parallel_size=1,
        pipeline_num_layers=llama_cfg.hidden_layers,
    ),
)
