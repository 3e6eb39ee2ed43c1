"""RoBERTa-base dynamic-MLM pretraining, synthetic data.

Reference recipe: configs/roberta_pretrain.py (no NSP, dynamic masking).
"""

import torch

from libai_amd.config import ConfigDict, LazyCall
from libai_amd.data import build_nlp_train_loader
from libai_amd.data.datasets import SyntheticBertDataset
from libai_amd.data.structures import DistTensorData, Instance
from libai_amd.models import RobertaForPreTraining

from .common.optim import optim
from .common.train import train

roberta_cfg = ConfigDict(
    vocab_size=50265,
    hidden_size=768,
    hidden_layers=12,
    num_attention_heads=12,
    intermediate_size=3072,
    max_position_embeddings=514,
)

model = LazyCall(RobertaForPreTraining)(cfg=roberta_cfg)


class _SyntheticRoberta(SyntheticBertDataset):
    """MLM-only samples (no NSP) shaped like RobertaDataset's output."""

    def __getitem__(self, idx):
        inst = super().__getitem__(idx)
        return Instance(
            input_ids=inst.input_ids,
            attention_mask=inst.attention_mask,
            tokentype_ids=DistTensorData(
                torch.zeros_like(inst.tokentype_ids.tensor)
            ),
            lm_labels=inst.lm_labels,
            loss_mask=inst.loss_mask,
        )


dataloader = dict(
    train=LazyCall(build_nlp_train_loader)(
        dataset=LazyCall(_SyntheticRoberta)(vocab_size=50265, seq_length=512,
                                            size=65536),
        train_batch_size=16,
        num_workers=2,
    ),
)

train.update(
    output_dir="./output/roberta_pretrain",
    train_micro_batch_size=16,
    train_iter=1000,
    log_period=10,
    amp=dict(enabled=True),
    dist=dict(
        data_parallel_size=None,
        tensor_parallel_size=1,
        pipeline_parallel_size=1,
        pipeline_num_layers=roberta_cfg.hidden_layers,
    ),
)
