"""T5 span-corruption dataset over an indexed corpus.

Reference behavior: libai/data/datasets/t5_dataset.py — sample token blocks
(build_blocks_mapping), corrupt ~15% of tokens in sentinel-marked spans
(mean length 3), produce encoder inputs with <extra_id_k> sentinels and
decoder targets listing the dropped spans.
"""

import numpy as np
import torch

from ..structures import DistTensorData, Instance

__all__ = ["T5Dataset", "build_span_corruption"]


def build_span_corruption(tokens, rng, sentinel_base, noise_density=0.15,
                          mean_span_length=3):
    """Return (enc_tokens, dec_input, dec_target) with sentinel ids counting
    DOWN from sentinel_base (the T5 <extra_id_0> convention)."""
    tokens = np.asarray(tokens, dtype=np.int64)
    n = len(tokens)
    n_noise = max(1, int(round(n * noise_density)))
    n_spans = max(1, int(round(n_noise / mean_span_length)))

    # choose span starts/lengths
    starts = np.sort(rng.choice(np.arange(1, max(n - 1, 2)),
                                size=min(n_spans, max(n // 4, 1)), replace=False))
    spans = []
    used = np.zeros(n, dtype=bool)
    per = max(1, n_noise // max(len(starts), 1))
    for s in starts:
        e = min(n, s + per)
        if used[s:e].any():
            continue
        used[s:e] = True
        spans.append((int(s), int(e)))
    if not spans:
        spans = [(0, min(per, n))]

    enc, dec = [], []
    pos = 0
    for k, (s, e) in enumerate(spans):
        sentinel = sentinel_base - k
        enc.extend(tokens[pos:s].tolist())
        enc.append(sentinel)
        dec.append(sentinel)
        dec.extend(tokens[s:e].tolist())
        pos = e
    enc.extend(tokens[pos:].tolist())
    dec.append(sentinel_base - len(spans))  # closing sentinel
    return (np.asarray(enc, dtype=np.int64), np.asarray(dec[:-1], dtype=np.int64),
            np.asarray(dec[1:] + [0], dtype=np.int64))


class T5Dataset(torch.utils.data.Dataset):
    def __init__(self, name, indexed_dataset, max_seq_length=512,
                 max_seq_length_dec=128, noise_density=0.15, mean_span_length=3,
                 num_samples=None, seed=1234, vocab_size=30522, pad_id=0,
                 sentinel_base=None):
        self.ds = indexed_dataset
        self.L_enc = max_seq_length
        self.L_dec = max_seq_length_dec
        self.noise = noise_density
        self.mean_span = mean_span_length
        self.seed = seed
        self.pad_id = pad_id
        self.sentinel_base = sentinel_base or (vocab_size - 1)

        docs = np.asarray(self.ds.doc_idx, dtype=np.int64)
        sizes = np.asarray(self.ds.sizes, dtype=np.int32)
        try:
            from libai_amd import _data_helpers

            self.mapping = np.asarray(
                _data_helpers.build_blocks_mapping(
                    docs, sizes, np.zeros(len(docs) - 1, dtype=np.int32), 1,
                    num_samples if num_samples is not None else (1 << 62),
                    max_seq_length - 2, seed, False, False,
                )
            )
        except ImportError:
            rows = []
            for d in range(len(docs) - 1):
                s0, s1 = docs[d], docs[d + 1]
                sent = s0
                while sent < s1:
                    end, tok = sent, 0
                    while end < s1 and tok + sizes[end] <= max_seq_length - 2:
                        tok += sizes[end]
                        end += 1
                    if end == sent:
                        end = sent + 1
                    rows.append((sent, end, d, max_seq_length - 2))
                    sent = end
            self.mapping = np.asarray(rows, dtype=np.int64)
        if num_samples is not None:
            self.mapping = self.mapping[:num_samples]

    def __len__(self):
        return len(self.mapping)

    def _pad(self, arr, L):
        arr = arr[:L]
        mask = np.concatenate([np.ones(len(arr), dtype=np.uint8),
                               np.zeros(L - len(arr), dtype=np.uint8)])
        out = np.concatenate([arr, np.full(L - len(arr), self.pad_id,
                                           dtype=np.int64)])
        return out, mask

    def __getitem__(self, idx):
        start, end = int(self.mapping[idx][0]), int(self.mapping[idx][1])
        rng = np.random.default_rng(self.seed + idx)
        tokens = np.concatenate(
            [np.asarray(self.ds[i], dtype=np.int64) for i in range(start, end)]
        )[: self.L_enc - 2]
        enc, dec_in, dec_tgt = build_span_corruption(
            tokens, rng, self.sentinel_base, self.noise, self.mean_span
        )
        enc, enc_mask = self._pad(enc, self.L_enc)
        dec_in, dec_mask = self._pad(dec_in, self.L_dec)
        dec_tgt, tgt_mask = self._pad(dec_tgt, self.L_dec)
        loss_mask = tgt_mask.astype(np.int64)
        return Instance(
            encoder_input_ids=DistTensorData(torch.from_numpy(enc)),
            decoder_input_ids=DistTensorData(torch.from_numpy(dec_in)),
            encoder_attn_mask=DistTensorData(torch.from_numpy(enc_mask)),
            decoder_attn_mask=DistTensorData(torch.from_numpy(dec_mask)),
            lm_labels=DistTensorData(torch.from_numpy(dec_tgt), placement_idx=-1),
            loss_mask=DistTensorData(torch.from_numpy(loss_mask), placement_idx=-1),
        )
