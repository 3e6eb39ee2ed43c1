"""Synthetic datasets for benchmarking and tests (random tokens / images).

The bench contract (BASELINE.json) runs on synthetic data with random-init
weights; these datasets generate deterministic per-index samples so loss
curves are reproducible across ranks and resumes.
"""

import torch

from ..structures import DistTensorData, Instance

__all__ = ["SyntheticGPTDataset", "SyntheticBertDataset", "SyntheticImageDataset"]


class SyntheticGPTDataset(torch.utils.data.Dataset):
    """GPT sample contract (reference: libai/data/datasets/gpt_dataset.py:92-98):
    Instance(input_ids [s], labels [s] tagged for the last stage)."""

    def __init__(self, vocab_size=50257, seq_length=1024, size=65536, seed=0):
        self.vocab_size = vocab_size
        self.seq_length = seq_length
        self.size = size
        self.seed = seed

    def __len__(self):
        return self.size

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + idx)
        tokens = torch.randint(0, self.vocab_size, (self.seq_length + 1,), generator=g)
        return Instance(
            input_ids=DistTensorData(tokens[:-1].long()),
            labels=DistTensorData(tokens[1:].long(), placement_idx=-1),
        )


class SyntheticBertDataset(torch.utils.data.Dataset):
    """BERT sample contract (reference: bert_dataset.py): masked-LM inputs."""

    def __init__(self, vocab_size=30522, seq_length=512, size=65536, seed=0,
                 mask_prob=0.15, mask_token_id=103):
        self.vocab_size = vocab_size
        self.seq_length = seq_length
        self.size = size
        self.seed = seed
        self.mask_prob = mask_prob
        self.mask_token_id = mask_token_id

    def __len__(self):
        return self.size

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + idx)
        tokens = torch.randint(5, self.vocab_size, (self.seq_length,), generator=g)
        mask = torch.rand(self.seq_length, generator=g) < self.mask_prob
        input_ids = tokens.clone()
        input_ids[mask] = self.mask_token_id
        lm_labels = torch.where(mask, tokens, torch.full_like(tokens, -1))
        ns_label = torch.randint(0, 2, (1,), generator=g)[0]
        return Instance(
            input_ids=DistTensorData(input_ids.long()),
            attention_mask=DistTensorData(torch.ones(self.seq_length, dtype=torch.uint8)),
            tokentype_ids=DistTensorData(torch.zeros(self.seq_length, dtype=torch.long)),
            ns_labels=DistTensorData(ns_label.long(), placement_idx=-1),
            lm_labels=DistTensorData(lm_labels.long(), placement_idx=-1),
            loss_mask=DistTensorData(mask.long(), placement_idx=-1),
        )


class SyntheticImageDataset(torch.utils.data.Dataset):
    """CV sample contract (reference: datasets/imagenet.py): images + labels."""

    def __init__(self, img_size=224, num_classes=1000, size=65536, seed=0):
        self.img_size = img_size
        self.num_classes = num_classes
        self.size = size
        self.seed = seed

    def __len__(self):
        return self.size

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed + idx)
        img = torch.randn(3, self.img_size, self.img_size, generator=g)
        label = torch.randint(0, self.num_classes, (1,), generator=g)[0]
        return Instance(
            images=DistTensorData(img),
            labels=DistTensorData(label.long(), placement_idx=-1),
        )
