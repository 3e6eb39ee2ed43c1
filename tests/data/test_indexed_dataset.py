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


def _sentence_corpus(tmp_path):
    prefix = str(tmp_path / "sents")
    b = MMapIndexedDatasetBuilder(data_file_path(prefix), np.uint16)
    rng = np.random.RandomState(0)
    for d in range(8):  # 8 docs x 5 sentences
        for s in range(5):
            b.add_item(rng.randint(200, 5000, size=rng.randint(5, 20)).astype(np.uint16))
        b.end_document()
    b.finalize(index_file_path(prefix))
    return MMapIndexedDataset(prefix)


def test_bert_dataset_masked_lm(tmp_path):
    from libai_amd.data.datasets import BertDataset

    ds = _sentence_corpus(tmp_path)
    bd = BertDataset("t", ds, max_seq_length=64, vocab_size=6000, num_samples=10)
    assert len(bd) > 0
    inst = bd[0]
    ids = inst.get("input_ids").tensor
    labels = inst.get("lm_labels").tensor
    loss_mask = inst.get("loss_mask").tensor
    assert ids.shape == (64,)
    assert ids[0] == 101  # [CLS]
    masked_positions = (loss_mask == 1).nonzero().flatten()
    assert len(masked_positions) >= 1
    # labels hold the ORIGINAL token at masked positions, -1 elsewhere
    assert (labels[loss_mask == 0] == -1).all()
    assert (labels[loss_mask == 1] >= 0).all()
    # deterministic per index
    inst2 = bd[0]
    assert torch.equal(ids, inst2.get("input_ids").tensor)


def test_t5_dataset_span_corruption(tmp_path):
    from libai_amd.data.datasets import T5Dataset

    ds = _sentence_corpus(tmp_path)
    td = T5Dataset("t", ds, max_seq_length=64, max_seq_length_dec=32,
                   vocab_size=6000, num_samples=8)
    assert len(td) > 0
    inst = td[0]
    enc = inst.get("encoder_input_ids").tensor
    dec_in = inst.get("decoder_input_ids").tensor
    tgt = inst.get("lm_labels").tensor
    assert enc.shape == (64,) and dec_in.shape == (32,) and tgt.shape == (32,)
    # sentinels (counting down from vocab-1) appear in encoder and decoder
    assert (enc >= 5900).any()
    assert (dec_in >= 5900).any()


def test_roberta_dataset_no_nsp(tmp_path):
    from libai_amd.data.datasets import RobertaDataset

    ds = _sentence_corpus(tmp_path)
    rd = RobertaDataset("t", ds, max_seq_length=64, vocab_size=6000,
                        num_samples=10)
    assert len(rd) > 0
    inst = rd[0]
    assert "ns_labels" not in inst.get_fields()
    assert (inst.get("tokentype_ids").tensor == 0).all()
    loss_mask = inst.get("loss_mask").tensor
    labels = inst.get("lm_labels").tensor
    assert (labels[loss_mask == 0] == -1).all()
    assert (labels[loss_mask == 1] >= 0).all()


def test_train_val_test_split_loader(tmp_path):
    """One corpus -> doc-range train/val/test datasets (reference build.py:28-150)."""
    from libai_amd.data.build import build_nlp_train_val_test_loader

    prefix = str(tmp_path / "corpus")
    b = MMapIndexedDatasetBuilder(data_file_path(prefix), np.uint16)
    rng = np.random.RandomState(0)
    for d in range(40):
        b.add_item(rng.randint(10, 5000, size=200).astype(np.uint16))
        b.end_document()
    b.finalize(index_file_path(prefix))

    train, valid, test = build_nlp_train_val_test_loader(
        prefix, splits="80,15,5", max_seq_length=64, train_batch_size=2,
        test_batch_size=2, num_workers=0,
    )
    batch = next(iter(train))
    ids = batch.get("input_ids").tensor
    assert ids.shape == (2, 64)
    # val/test datasets see disjoint, smaller doc ranges
    assert valid is not None and test is not None
    n_tr = len(train.batch_sampler.dataset.documents) if hasattr(
        train.batch_sampler, "dataset") else None
    vb = next(iter(valid))
    assert vb.get("input_ids").tensor.shape[1] == 64


def test_blendable_dataset_weights():
    from libai_amd.data import BlendableDataset

    class _Fixed(torch.utils.data.Dataset):
        def __init__(self, tag, n=100):
            self.tag, self.n = tag, n

        def __len__(self):
            return self.n

        def __getitem__(self, i):
            return (self.tag, i)

    bd = BlendableDataset([_Fixed("a"), _Fixed("b")], weights=[0.7, 0.3],
                          size=1000)
    tags = [bd[i][0] for i in range(1000)]
    frac_a = tags.count("a") / 1000
    assert abs(frac_a - 0.7) < 0.02
    # per-dataset sample indices advance sequentially (modulo wrap)
    seen_a = [bd[i][1] for i in range(50) if bd[i][0] == "a"]
    assert seen_a == sorted(seen_a)
