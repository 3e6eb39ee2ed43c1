import itertools

import torch

from libai_amd.data import CyclicSampler, SingleRoundSampler


class _Range(torch.utils.data.Dataset):
    def __init__(self, n):
        self.n = n

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        return i


def test_cyclic_sampler_disjoint_dp_shards():
    ds = _Range(32)
    s0 = CyclicSampler(ds, 4, data_parallel_rank=0, data_parallel_size=2)
    s1 = CyclicSampler(ds, 4, data_parallel_rank=1, data_parallel_size=2)
    b0 = list(itertools.islice(iter(s0), 4))
    b1 = list(itertools.islice(iter(s1), 4))
    for a, b in zip(b0, b1):
        assert len(a) == len(b) == 4
        assert not (set(a) & set(b)), "dp ranks saw overlapping samples"


def test_cyclic_sampler_cycles_epochs():
    ds = _Range(8)
    s = CyclicSampler(ds, 4)
    batches = list(itertools.islice(iter(s), 5))
    assert batches[0] == batches[2]  # epoch of 2 batches repeats (no shuffle)


def test_cyclic_sampler_resume_via_consumed_samples():
    ds = _Range(64)
    s = CyclicSampler(ds, 4, shuffle=True, seed=3)
    first = list(itertools.islice(iter(s), 6))
    consumed_after_3 = 3 * 4  # 3 batches x global batch 4
    s2 = CyclicSampler(ds, 4, shuffle=True, seed=3, consumed_samples=consumed_after_3)
    resumed = list(itertools.islice(iter(s2), 3))
    assert resumed == first[3:6], "resume must continue the same stream"


def test_single_round_sampler_covers_all_once():
    ds = _Range(10)
    s = SingleRoundSampler(ds, 3, data_parallel_rank=0, data_parallel_size=1)
    out = [i for batch in s for i in batch]
    assert sorted(set(out)) == list(range(10))
    assert len(out) == 12  # padded to batch multiple


def test_single_round_sampler_drop_last():
    ds = _Range(10)
    s = SingleRoundSampler(ds, 3, drop_last=True)
    out = [i for batch in s for i in batch]
    assert len(out) == 9
