import numpy as np
import pytest
import torch

from libai_amd.data.indexed_dataset import (
    MMapIndexedDataset,
    MMapIndexedDatasetBuilder,
    best_fitting_dtype,
    data_file_path,
    index_file_path,
)


@pytest.fixture()
def corpus(tmp_path):
    prefix = str(tmp_path / "corpus")
    b = MMapIndexedDatasetBuilder(data_file_path(prefix), np.uint16)
    docs = [
        np.arange(10, dtype=np.uint16),
        np.arange(100, 125, dtype=np.uint16),
        np.arange(7, dtype=np.uint16) + 1000,
    ]
    for d in docs:
        b.add_item(d)
        b.end_document()
    b.finalize(index_file_path(prefix))
    return prefix, docs


def test_roundtrip(corpus):
    prefix, docs = corpus
    ds = MMapIndexedDataset(prefix)
    assert len(ds) == 3
    for i, d in enumerate(docs):
        assert np.array_equal(ds[i], d)
    assert np.array_equal(ds.get(1, offset=5, length=3), docs[1][5:8])
    assert list(ds.doc_idx) == [0, 1, 2, 3]


def test_best_fitting_dtype():
    assert best_fitting_dtype(50000) == np.uint16
    assert best_fitting_dtype(100000) == np.int32


def test_gpt2_dataset_chopping(corpus):
    from libai_amd.data.datasets.gpt_dataset import GPT2Dataset

    prefix, docs = corpus
    ds = MMapIndexedDataset(prefix)
    g = GPT2Dataset("test", ds, max_seq_length=8, num_samples=6, seed=0)
    assert len(g) == 6
    for i in range(len(g)):
        inst = g[i]
        ids = inst.get("input_ids").tensor
        labels = inst.get("labels").tensor
        assert ids.shape == (8,) and labels.shape == (8,)
        # next-token alignment
        assert torch.equal(ids[1:], labels[:-1])


def test_cpp_helpers_match_numpy_fallback(corpus):
    helpers = pytest.importorskip("libai_amd._data_helpers")
    from libai_amd.data.datasets.gpt_dataset import _build_sample_idx_np

    prefix, docs = corpus
    ds = MMapIndexedDataset(prefix)
    sizes = ds.sizes.astype(np.int32)
    doc_idx = np.array([0, 1, 2, 0, 1, 2], dtype=np.int32)
    tokens = int(sizes.sum())
    a = helpers.build_sample_idx(sizes, doc_idx, 8, 2, tokens)
    b = _build_sample_idx_np(sizes, doc_idx, 8, 2, tokens)
    assert np.array_equal(np.asarray(a), b)


def test_blending_indices():
    helpers = pytest.importorskip("libai_amd._data_helpers")
    w = np.array([0.7, 0.3])
    di, dsi = helpers.build_blending_indices(w, 1000)
    frac = np.bincount(np.asarray(di), minlength=2) / 1000
    assert abs(frac[0] - 0.7) < 0.01
    # per-dataset sample indices are sequential
    for d in (0, 1):
        sel = np.asarray(dsi)[np.asarray(di) == d]
        assert np.array_equal(sel, np.arange(len(sel)))
