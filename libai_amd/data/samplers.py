"""Data samplers (reference: libai/data/samplers/samplers.py:20-186).

CyclicSampler: infinite epoch-cycled stream of per-DP-rank micro-batch index
lists, resumable via ``consumed_samples``.  SingleRoundSampler: one pass for
eval, padding the uneven tail so every DP rank sees the same batch count.
"""

import torch
from torch.utils.data import Sampler

__all__ = ["CyclicSampler", "SingleRoundSampler"]


class CyclicSampler(Sampler):
    def __init__(self, dataset, micro_batch_size, shuffle=False, consumed_samples=0,
                 data_parallel_rank=0, data_parallel_size=1, seed=0):
        self.dataset = dataset
        self.data_size = len(dataset)
        self.shuffle = shuffle
        self.micro_batch_size = micro_batch_size
        self.dp_rank = data_parallel_rank
        self.dp_size = data_parallel_size
        self.actual_batch_size = micro_batch_size * data_parallel_size
        self.data_size_per_epoch = self.data_size // self.actual_batch_size * self.actual_batch_size
        self.consumed_samples = consumed_samples
        self.seed = seed

    def __iter__(self):
        """Yield lists of micro_batch_size indices for THIS dp rank, forever."""
        epoch = self.consumed_samples // self.data_size_per_epoch
        current = self.consumed_samples % self.data_size_per_epoch
        batch = []
        while True:
            current = current % self.data_size_per_epoch
            indices = self._get_epoch_indices(epoch)
            # stride the global batch across dp ranks
            start = current + self.dp_rank * self.micro_batch_size
            for i in range(start, self.data_size_per_epoch, self.actual_batch_size):
                for j in range(self.micro_batch_size):
                    if i + j < self.data_size_per_epoch:
                        batch.append(int(indices[i + j]))
                    if len(batch) == self.micro_batch_size:
                        self.consumed_samples += self.actual_batch_size
                        yield batch
                        batch = []
            epoch += 1
            current = 0

    def _get_epoch_indices(self, epoch):
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed + epoch)
            return torch.randperm(self.data_size_per_epoch, generator=g).tolist()
        return list(range(self.data_size_per_epoch))

    def set_consumed_samples(self, consumed_samples):
        self.consumed_samples = consumed_samples

    def set_epoch(self, epoch):
        self.epoch = epoch


class SingleRoundSampler(Sampler):
    """One pass over the dataset, padded so every dp rank yields equally
    (reference: samplers.py:109-186)."""

    def __init__(self, dataset, micro_batch_size, shuffle=False, data_parallel_rank=0,
                 data_parallel_size=1, seed=0, drop_last=False):
        self.dataset = dataset
        self.data_size = len(dataset)
        self.shuffle = shuffle
        self.micro_batch_size = micro_batch_size
        self.dp_rank = data_parallel_rank
        self.dp_size = data_parallel_size
        self.seed = seed
        self.drop_last = drop_last

    def __iter__(self):
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed)
            indices = torch.randperm(self.data_size, generator=g).tolist()
        else:
            indices = list(range(self.data_size))
        global_batch = self.micro_batch_size * self.dp_size
        if not self.drop_last and len(indices) % global_batch != 0:
            pad = global_batch - len(indices) % global_batch
            indices = indices + indices[:pad]
        elif self.drop_last:
            indices = indices[: len(indices) // global_batch * global_batch]
        batch = []
        for i in range(self.dp_rank * self.micro_batch_size, len(indices), global_batch):
            chunk = indices[i : i + self.micro_batch_size]
            if len(chunk) == self.micro_batch_size:
                yield chunk

    def __len__(self):
        global_batch = self.micro_batch_size * self.dp_size
        if self.drop_last:
            return self.data_size // global_batch
        return (self.data_size + global_batch - 1) // global_batch
