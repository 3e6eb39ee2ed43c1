"""Dataloader builders (reference: libai/data/build.py:28-401)."""

import torch
from torch.utils.data import ConcatDataset, DataLoader

from ..utils import distributed as du
from .samplers import CyclicSampler, SingleRoundSampler
from .structures import Instance

__all__ = [
    "build_nlp_train_loader",
    "build_nlp_test_loader",
    "build_image_train_loader",
    "build_image_test_loader",
    "trivial_batch_collator",
    "build_train_valid_test_loaders",
    "build_nlp_train_val_test_loader",
]


def trivial_batch_collator(batch):
    """List[Instance] -> batched Instance (reference: build.py:398+)."""
    assert isinstance(batch[0], Instance), "dataset must return Instance objects"
    return Instance.stack(batch)


def _train_loader(dataset, micro_batch_size, shuffle, consumed_samples, seed,
                  num_workers, collate_fn):
    dutil = du.get_dist_util()
    sampler = CyclicSampler(
        dataset,
        micro_batch_size,
        shuffle=shuffle,
        consumed_samples=consumed_samples,
        data_parallel_rank=dutil.data_parallel_rank,
        data_parallel_size=dutil.data_parallel_size,
        seed=seed,
    )
    return DataLoader(
        dataset,
        batch_sampler=sampler,
        num_workers=num_workers,
        collate_fn=collate_fn or trivial_batch_collator,
        pin_memory=torch.cuda.is_available(),
        persistent_workers=num_workers > 0,
    )


def build_nlp_train_loader(dataset, train_batch_size, test_batch_size=None,
                           sampler=None, num_workers=4, consumed_samples=0, seed=0,
                           collate_fn=None, dataset_mixer=None, weights=None,
                           shuffle=True, **kwargs):
    if isinstance(dataset, (list, tuple)):
        if len(dataset) == 1:
            dataset = dataset[0]
        elif weights is not None or dataset_mixer is not None:
            # weighted multi-corpus blending (reference: blendable dataset
            # over build_blending_indices)
            from .blendable import BlendableDataset

            mixer = dataset_mixer or BlendableDataset
            dataset = mixer(dataset, weights or [1.0] * len(dataset))
        else:
            dataset = ConcatDataset(dataset)
    loader = _train_loader(
        dataset, train_batch_size, shuffle, consumed_samples, seed, num_workers,
        collate_fn,
    )
    return loader, None, None


def build_nlp_test_loader(dataset, test_batch_size, sampler=None, num_workers=4,
                          seed=0, collate_fn=None, **kwargs):
    dutil = du.get_dist_util()
    sampler = sampler or SingleRoundSampler(
        dataset,
        test_batch_size,
        shuffle=False,
        data_parallel_rank=dutil.data_parallel_rank,
        data_parallel_size=dutil.data_parallel_size,
        seed=seed,
    )
    return DataLoader(
        dataset,
        batch_sampler=sampler,
        num_workers=num_workers,
        collate_fn=collate_fn or trivial_batch_collator,
        pin_memory=torch.cuda.is_available(),
    )


def build_image_train_loader(dataset, train_batch_size, test_batch_size=None,
                             sampler=None, num_workers=4, consumed_samples=0, seed=0,
                             collate_fn=None, mixup_func=None, shuffle=True, **kwargs):
    if isinstance(dataset, (list, tuple)):
        dataset = dataset[0] if len(dataset) == 1 else ConcatDataset(dataset)
    loader = _train_loader(
        dataset, train_batch_size, shuffle, consumed_samples, seed, num_workers,
        collate_fn,
    )
    loader.mixup_func = mixup_func
    return loader, None, None


def build_image_test_loader(dataset, test_batch_size, sampler=None, num_workers=4,
                            seed=0, collate_fn=None, **kwargs):
    return build_nlp_test_loader(dataset, test_batch_size, sampler, num_workers, seed,
                                 collate_fn, **kwargs)


def build_train_valid_test_loaders(train_dataset, valid_dataset, test_dataset,
                                   train_batch_size, test_batch_size,
                                   num_workers=4, consumed_samples=0, seed=0,
                                   collate_fn=None):
    """Reference build_nlp_train_val_test_loader analog (build.py:28-150)."""
    train, _, _ = build_nlp_train_loader(
        train_dataset, train_batch_size, num_workers=num_workers,
        consumed_samples=consumed_samples, seed=seed, collate_fn=collate_fn,
    )
    valid = (
        build_nlp_test_loader(valid_dataset, test_batch_size, num_workers=num_workers,
                              collate_fn=collate_fn)
        if valid_dataset is not None
        else None
    )
    test = (
        build_nlp_test_loader(test_dataset, test_batch_size, num_workers=num_workers,
                              collate_fn=collate_fn)
        if test_dataset is not None
        else None
    )
    return train, valid, test


def build_nlp_train_val_test_loader(data_prefix, splits="949,50,1",
                                    max_seq_length=1024,
                                    train_val_test_num_samples=(None, None, None),
                                    train_batch_size=4, test_batch_size=4,
                                    num_workers=4, consumed_samples=0, seed=1234,
                                    dataset_cls=None, collate_fn=None):
    """Split ONE indexed corpus into train/val/test by document ranges
    (reference: libai/data/build.py:28-150).

    ``splits`` is the Megatron-style comma ratio string ("949,50,1"); the
    document index space of ``data_prefix``'s .idx/.bin pair is partitioned
    proportionally and each partition backs its own GPT2Dataset.
    """
    import numpy as np

    from .datasets.gpt_dataset import GPT2Dataset
    from .indexed_dataset import MMapIndexedDataset

    dataset_cls = dataset_cls or GPT2Dataset
    indexed = MMapIndexedDataset(data_prefix)
    n_docs = len(indexed.doc_idx) - 1
    ratios = [float(x) for x in str(splits).split(",")]
    while len(ratios) < 3:
        ratios.append(0.0)
    total = sum(ratios[:3])
    bounds = [0]
    acc = 0.0
    for r in ratios[:3]:
        acc += r
        bounds.append(int(round(n_docs * acc / total)))
    bounds[-1] = n_docs

    names = ("train", "valid", "test")
    datasets = []
    for i, name in enumerate(names):
        lo, hi = bounds[i], bounds[i + 1]
        if hi <= lo:
            datasets.append(None)
            continue
        datasets.append(dataset_cls(
            name, indexed, documents=np.arange(lo, hi, dtype=np.int32),
            num_samples=train_val_test_num_samples[i],
            max_seq_length=max_seq_length, seed=seed,
        ))
    return build_train_valid_test_loaders(
        datasets[0], datasets[1], datasets[2], train_batch_size,
        test_batch_size, num_workers=num_workers,
        consumed_samples=consumed_samples, seed=seed, collate_fn=collate_fn,
    )
