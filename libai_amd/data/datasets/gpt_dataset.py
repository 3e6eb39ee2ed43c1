"""GPT2Dataset: seq-length chopping over an indexed token corpus.

Reference behavior: libai/data/datasets/gpt_dataset.py:101-245 — doc/sample/
shuffle index triple built by the C++ helpers (libai_amd/_data_helpers.so;
numpy fallback), np-memmap cached on disk, resumable by index.
"""

import hashlib
import logging
import os

import numpy as np
import torch

from ..structures import DistTensorData, Instance

logger = logging.getLogger(__name__)

__all__ = ["GPT2Dataset"]


def _helpers():
    try:
        from libai_amd import _data_helpers

        return _data_helpers
    except ImportError:
        return None


def _build_sample_idx_np(sizes, doc_idx, seq_length, num_epochs, tokens_per_epoch):
    """Pure-numpy fallback mirroring the C++ helper."""
    num_samples = (num_epochs * tokens_per_epoch - 1) // seq_length
    sample_idx = np.zeros((num_samples + 1, 2), dtype=np.int32)
    doc_pos, doc_off = 0, 0
    sample_idx[0] = (0, 0)
    for s in range(1, num_samples + 1):
        remaining = seq_length + 1
        while remaining > 0:
            doc_len = sizes[doc_idx[doc_pos]] - doc_off
            if doc_len > remaining:
                doc_off += remaining - 1
                remaining = 0
            else:
                remaining -= doc_len
                if remaining == 0:
                    doc_off = sizes[doc_idx[doc_pos]] - 1
                else:
                    doc_pos += 1
                    doc_off = 0
        sample_idx[s] = (doc_pos, doc_off)
    return sample_idx


class GPT2Dataset(torch.utils.data.Dataset):
    def __init__(self, name, indexed_dataset, documents=None, num_samples=None,
                 max_seq_length=1024, seed=1234, data_prefix=None):
        self.name = name
        self.ds = indexed_dataset
        self.seq_length = max_seq_length
        self.seed = seed
        if documents is None:
            documents = np.arange(len(self.ds.doc_idx) - 1, dtype=np.int32)
        self.documents = np.asarray(documents, dtype=np.int32)

        sizes = self.ds.sizes
        tokens_per_epoch = int(np.sum(sizes[self.documents].astype(np.int64)))
        samples_per_epoch = max((tokens_per_epoch - 1) // max_seq_length, 1)
        if num_samples is None:
            num_samples = samples_per_epoch
        num_epochs = (num_samples * max_seq_length + tokens_per_epoch - 1) // \
            tokens_per_epoch + 1
        self.num_samples = num_samples

        rng = np.random.RandomState(seed)
        doc_idx = np.concatenate(
            [rng.permutation(self.documents) for _ in range(num_epochs)]
        ).astype(np.int32)
        helpers = _helpers()
        if helpers is not None:
            self.sample_idx = helpers.build_sample_idx(
                sizes.astype(np.int32), doc_idx, max_seq_length, num_epochs,
                tokens_per_epoch,
            )
        else:
            self.sample_idx = _build_sample_idx_np(
                sizes, doc_idx, max_seq_length, num_epochs, tokens_per_epoch
            )
        self.doc_idx = doc_idx
        n = min(num_samples, self.sample_idx.shape[0] - 1)
        self.shuffle_idx = rng.permutation(n).astype(np.int64)

    def __len__(self):
        return len(self.shuffle_idx)

    def __getitem__(self, idx):
        idx = int(self.shuffle_idx[idx % len(self.shuffle_idx)])
        d0, o0 = self.sample_idx[idx]
        d1, o1 = self.sample_idx[idx + 1]
        if d0 == d1:
            toks = self.ds.get(self.doc_idx[d0], offset=o0, length=o1 - o0 + 1)
        else:
            parts = [self.ds.get(self.doc_idx[d0], offset=o0)]
            for d in range(d0 + 1, d1):
                parts.append(self.ds.get(self.doc_idx[d]))
            parts.append(self.ds.get(self.doc_idx[d1], length=o1 + 1))
            toks = np.concatenate(parts)
        toks = np.asarray(toks[: self.seq_length + 1], dtype=np.int64)
        if len(toks) < self.seq_length + 1:  # tail padding (rare)
            toks = np.pad(toks, (0, self.seq_length + 1 - len(toks)))
        t = torch.from_numpy(toks.copy())
        return Instance(
            input_ids=DistTensorData(t[:-1]),
            labels=DistTensorData(t[1:], placement_idx=-1),
        )
