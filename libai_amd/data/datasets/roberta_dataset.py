"""RoBERTa masked-LM dataset: BERT-style span sampling without NSP.

Reference behavior: libai/data/datasets/roberta_dataset.py — identical
sentence-range mapping and 15% dynamic masking as the BERT dataset, but a
single contiguous segment ([CLS] A [SEP], tokentype all zeros) and no
next-sentence label; masking is re-drawn per epoch visit (dynamic masking).
"""

import torch

from ..structures import DistTensorData, Instance
from .bert_dataset import BertDataset

__all__ = ["RobertaDataset"]


class RobertaDataset(BertDataset):
    def __init__(self, *args, **kwargs):
        kwargs["binary_head"] = False
        super().__init__(*args, **kwargs)

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
