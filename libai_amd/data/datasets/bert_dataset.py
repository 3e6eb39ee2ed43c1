"""BERT masked-LM + next-sentence dataset over an indexed sentence corpus.

Reference behavior: libai/data/datasets/bert_dataset.py +
data_utils/dataset_utils.py:79-324 (create_masked_lm_predictions): sentence
ranges come from the build_mapping index helper; each sample is a
[CLS] A [SEP] B [SEP] pair (50% random B -> is_next=0) with 15% of tokens
masked (80% [MASK] / 10% random / 10% kept).
"""

import numpy as np
import torch

from ..structures import DistTensorData, Instance

__all__ = ["BertDataset", "create_masked_lm_predictions"]


def create_masked_lm_predictions(tokens, vocab_size, mask_id, rng,
                                 masked_lm_prob=0.15, special_ids=(),
                                 max_predictions=None):
    """Return (masked_tokens, lm_labels, loss_mask); labels -1 where unmasked."""
    tokens = np.asarray(tokens, dtype=np.int64).copy()
    labels = np.full_like(tokens, -1)
    loss_mask = np.zeros_like(tokens)
    special = set(int(s) for s in special_ids)
    cand = [i for i, t in enumerate(tokens) if int(t) not in special]
    rng.shuffle(cand)
    n_pred = max(1, int(round(len(cand) * masked_lm_prob)))
    if max_predictions is not None:
        n_pred = min(n_pred, max_predictions)
    for i in cand[:n_pred]:
        labels[i] = tokens[i]
        loss_mask[i] = 1
        r = rng.random()
        if r < 0.8:
            tokens[i] = mask_id
        elif r < 0.9:
            tokens[i] = rng.integers(0, vocab_size)
        # else: keep original
    return tokens, labels, loss_mask


class BertDataset(torch.utils.data.Dataset):
    def __init__(self, name, indexed_dataset, max_seq_length=512,
                 masked_lm_prob=0.15, num_samples=None, seed=1234,
                 cls_id=101, sep_id=102, mask_id=103, pad_id=0, vocab_size=30522,
                 short_seq_prob=0.1, binary_head=True):
        self.ds = indexed_dataset
        self.max_seq_length = max_seq_length
        self.masked_lm_prob = masked_lm_prob
        self.cls_id, self.sep_id, self.mask_id, self.pad_id = (cls_id, sep_id,
                                                               mask_id, pad_id)
        self.vocab_size = vocab_size
        self.seed = seed
        self.binary_head = binary_head

        docs = np.asarray(self.ds.doc_idx, dtype=np.int64)
        sizes = np.asarray(self.ds.sizes, dtype=np.int32)
        target = max_seq_length - 3  # [CLS] + 2x[SEP]
        num_epochs = 1
        if num_samples is not None:
            approx_per_epoch = max(len(sizes) // 4, 1)
            num_epochs = max(1, (num_samples + approx_per_epoch - 1) // approx_per_epoch)
        try:
            from libai_amd import _data_helpers

            self.mapping = np.asarray(
                _data_helpers.build_mapping(
                    docs, sizes, num_epochs,
                    num_samples if num_samples is not None else (1 << 62),
                    target, short_seq_prob, seed, False, 2 if binary_head else 1,
                )
            )
        except ImportError:
            self.mapping = self._build_mapping_np(docs, sizes, num_epochs, target,
                                                  short_seq_prob,
                                                  2 if binary_head else 1)
        if num_samples is not None:
            self.mapping = self.mapping[:num_samples]

    def _build_mapping_np(self, docs, sizes, num_epochs, target, short_seq_prob,
                          min_sent):
        rng = np.random.default_rng(self.seed)
        rows = []
        for _ in range(num_epochs):
            for d in range(len(docs) - 1):
                s0, s1 = docs[d], docs[d + 1]
                sent = s0
                while sent < s1:
                    tl = target
                    if rng.random() < short_seq_prob:
                        tl = 2 + int(rng.random() * (target - 2))
                    end, tok = sent, 0
                    while end < s1 and tok + sizes[end] <= tl:
                        tok += sizes[end]
                        end += 1
                    if end == sent:
                        end = sent + 1
                    if end - sent >= min_sent or end >= s1:
                        rows.append((sent, end, tl))
                    sent = end
        return np.asarray(rows, dtype=np.int64)

    def __len__(self):
        return len(self.mapping)

    def __getitem__(self, idx):
        start, end, target = self.mapping[idx]
        rng = np.random.default_rng(self.seed + idx)
        sents = [np.asarray(self.ds[int(i)], dtype=np.int64)
                 for i in range(int(start), int(end))]

        if self.binary_head and len(sents) > 1:
            split = 1 + int(rng.integers(0, len(sents) - 1))
            a = np.concatenate(sents[:split])
            b = np.concatenate(sents[split:])
            is_next = 1
            if rng.random() < 0.5:
                # random B from a random sample's tail
                j = int(rng.integers(0, len(self.mapping)))
                js, je, _ = self.mapping[j]
                b = np.concatenate(
                    [np.asarray(self.ds[int(i)], dtype=np.int64)
                     for i in range(int(js), int(je))]
                )
                is_next = 0
        else:
            a = np.concatenate(sents)
            b = np.array([], dtype=np.int64)
            is_next = 1

        max_tok = self.max_seq_length - 3
        while len(a) + len(b) > max_tok:
            if len(a) >= len(b):
                a = a[:-1]
            else:
                b = b[:-1]

        tokens = np.concatenate(
            [[self.cls_id], a, [self.sep_id], b, [self.sep_id]]
        )
        tokentype = np.concatenate(
            [np.zeros(len(a) + 2, dtype=np.int64),
             np.ones(len(b) + 1, dtype=np.int64)]
        )
        masked, labels, loss_mask = create_masked_lm_predictions(
            tokens, self.vocab_size, self.mask_id, rng,
            self.masked_lm_prob,
            special_ids=(self.cls_id, self.sep_id, self.pad_id),
        )
        # pad
        L = self.max_seq_length
        pad = L - len(masked)
        attn = np.concatenate([np.ones(len(masked), dtype=np.uint8),
                               np.zeros(pad, dtype=np.uint8)])
        masked = np.concatenate([masked, np.full(pad, self.pad_id, dtype=np.int64)])
        labels = np.concatenate([labels, np.full(pad, -1, dtype=np.int64)])
        loss_mask = np.concatenate([loss_mask, np.zeros(pad, dtype=np.int64)])
        tokentype = np.concatenate([tokentype, np.zeros(pad, dtype=np.int64)])

        return Instance(
            input_ids=DistTensorData(torch.from_numpy(masked)),
            attention_mask=DistTensorData(torch.from_numpy(attn)),
            tokentype_ids=DistTensorData(torch.from_numpy(tokentype)),
            ns_labels=DistTensorData(torch.tensor(is_next), placement_idx=-1),
            lm_labels=DistTensorData(torch.from_numpy(labels), placement_idx=-1),
            loss_mask=DistTensorData(torch.from_numpy(loss_mask), placement_idx=-1),
        )
