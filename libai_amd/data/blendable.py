"""Weighted multi-corpus blending (reference: Megatron-style blendable
dataset over build_blending_indices, helpers.cpp:34-84).

BlendableDataset presents N member datasets as one stream whose composition
follows the given weights; index maps come from the C++ helper (error-feedback
rounding keeps realized fractions within 1/size of the targets).
"""

import numpy as np
import torch

__all__ = ["BlendableDataset"]


class BlendableDataset(torch.utils.data.Dataset):
    def __init__(self, datasets, weights, size=None):
        assert len(datasets) == len(weights) and len(datasets) > 0
        self.datasets = list(datasets)
        w = np.asarray(weights, dtype=np.float64)
        w = w / w.sum()
        self.size = int(size) if size is not None else sum(
            len(d) for d in self.datasets
        )
        try:
            from libai_amd import _data_helpers

            di, dsi = _data_helpers.build_blending_indices(w, self.size)
            self.dataset_index = np.asarray(di, dtype=np.int64)
            self.dataset_sample_index = np.asarray(dsi, dtype=np.int64)
        except ImportError:  # pure-python fallback
            self.dataset_index = np.zeros(self.size, dtype=np.int64)
            self.dataset_sample_index = np.zeros(self.size, dtype=np.int64)
            counts = np.zeros(len(w))
            for i in range(self.size):
                errs = w * (i + 1) - counts
                d = int(np.argmax(errs))
                self.dataset_index[i] = d
                self.dataset_sample_index[i] = counts[d]
                counts[d] += 1

    def __len__(self):
        return self.size

    def __getitem__(self, idx):
        d = self.dataset_index[idx]
        s = self.dataset_sample_index[idx] % len(self.datasets[d])
        return self.datasets[d][int(s)]
