"""BERT-large pretraining, synthetic data (BASELINE config #2: bf16 DP8)."""

from libai_amd.config import LazyCall
from libai_amd.data import build_nlp_train_loader
from libai_amd.data.datasets import SyntheticBertDataset

from .common.models.bert import cfg as bert_cfg
from .common.models.bert import pretrain_model as model
from .common.optim import optim
from .common.train import train

# BERT-large: nl24, hidden 1024, 16 heads, seq 512 (Benchmark.md:12-26)
bert_cfg.hidden_size = 1024
bert_cfg.num_attention_heads = 16
bert_cfg.intermediate_size = 4096
bert_cfg.hidden_layers = 24

dataloader = dict(
    train=LazyCall(build_nlp_train_loader)(
        dataset=LazyCall(SyntheticBertDataset)(
            vocab_size=bert_cfg.vocab_size, seq_length=512, size=65536
        ),
        train_batch_size=16,
        num_workers=2,
    ),
)

train.update(
    output_dir="./output/bert_large_pretrain",
    train_micro_batch_size=16,
    train_iter=1000,
    log_period=10,
    amp=dict(enabled=True),
    dist=dict(
        data_parallel_size=None,
        tensor_parallel_size=1,
        pipeline_parallel_size=1,
        pipeline_num_layers=bert_cfg.hidden_layers,
    ),
)
