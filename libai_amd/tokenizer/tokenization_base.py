"""Tokenizer base class (reference: libai/tokenizer/tokenization_base.py).

A compact HF-v3-style PreTrainedTokenizer: vocab + special-token registry,
added-token handling, encode/decode protocol, and ``padded_vocab_size`` so
TP can require a divisible vocab (reference tokenization_base.py:489-494).
"""

import json
import logging
import os

__all__ = ["PreTrainedTokenizer"]

logger = logging.getLogger(__name__)

SPECIAL_TOKENS_MAP_FILE = "special_tokens_map.json"
ADDED_TOKENS_FILE = "added_tokens.json"


class PreTrainedTokenizer:
    SPECIAL_TOKENS_ATTRIBUTES = [
        "bos_token", "eos_token", "unk_token", "sep_token", "pad_token",
        "cls_token", "mask_token", "additional_special_tokens",
    ]
    vocab_files_names = {}

    def __init__(self, bos_token=None, eos_token=None, unk_token=None,
                 sep_token=None, pad_token=None, cls_token=None, mask_token=None,
                 additional_special_tokens=None, **kwargs):
        self.bos_token = bos_token
        self.eos_token = eos_token
        self.unk_token = unk_token
        self.sep_token = sep_token
        self.pad_token = pad_token
        self.cls_token = cls_token
        self.mask_token = mask_token
        self.additional_special_tokens = additional_special_tokens or []
        self.added_tokens_encoder = {}
        self.added_tokens_decoder = {}

    # -- vocab protocol (subclasses implement) ------------------------------

    @property
    def vocab_size(self):
        raise NotImplementedError

    def get_vocab(self):
        raise NotImplementedError

    def _tokenize(self, text):
        raise NotImplementedError

    def _convert_token_to_id(self, token):
        raise NotImplementedError

    def _convert_id_to_token(self, index):
        raise NotImplementedError

    def convert_tokens_to_string(self, tokens):
        return " ".join(tokens)

    # -- public API ---------------------------------------------------------

    def __len__(self):
        return self.vocab_size + len(self.added_tokens_encoder)

    def padded_vocab_size(self, multiple=1):
        """Vocab size rounded up so TP shards divide evenly (reference
        tokenization_base.py:489-494)."""
        size = len(self)
        while size % multiple != 0:
            size += 1
        return size

    @property
    def all_special_tokens(self):
        toks = []
        for attr in self.SPECIAL_TOKENS_ATTRIBUTES:
            v = getattr(self, attr, None)
            if v is None:
                continue
            if isinstance(v, (list, tuple)):
                toks.extend(v)
            else:
                toks.append(v)
        return toks

    @property
    def all_special_ids(self):
        return [self.convert_tokens_to_ids(t) for t in self.all_special_tokens]

    def add_tokens(self, new_tokens):
        added = 0
        for tok in new_tokens:
            if tok in self.added_tokens_encoder or self._convert_token_to_id(
                tok
            ) != self._convert_token_to_id(self.unk_token or ""):
                if tok in self.get_vocab():
                    continue
            idx = len(self)
            self.added_tokens_encoder[tok] = idx
            self.added_tokens_decoder[idx] = tok
            added += 1
        return added

    def add_special_tokens(self, special_tokens_dict):
        added = 0
        for key, value in special_tokens_dict.items():
            assert key in self.SPECIAL_TOKENS_ATTRIBUTES, f"unknown special {key}"
            setattr(self, key, value)
            if isinstance(value, (list, tuple)):
                added += self.add_tokens(value)
            else:
                added += self.add_tokens([value])
        return added

    def tokenize(self, text):
        # split on added/special tokens first, then subclass tokenization
        specials = [t for t in self.all_special_tokens if t] + list(
            self.added_tokens_encoder
        )
        if not specials:
            return self._tokenize(text)
        pieces = [text]
        for sp in specials:
            out = []
            for piece in pieces:
                if piece in specials:
                    out.append(piece)
                    continue
                parts = piece.split(sp)
                for i, part in enumerate(parts):
                    if part:
                        out.append(part)
                    if i < len(parts) - 1:
                        out.append(sp)
            pieces = out
        tokens = []
        for piece in pieces:
            if piece in specials:
                tokens.append(piece)
            else:
                tokens.extend(self._tokenize(piece))
        return tokens

    def convert_tokens_to_ids(self, tokens):
        if tokens is None:
            return None
        if isinstance(tokens, str):
            if tokens in self.added_tokens_encoder:
                return self.added_tokens_encoder[tokens]
            return self._convert_token_to_id(tokens)
        return [self.convert_tokens_to_ids(t) for t in tokens]

    def convert_ids_to_tokens(self, ids, skip_special_tokens=False):
        if isinstance(ids, int):
            if ids in self.added_tokens_decoder:
                return self.added_tokens_decoder[ids]
            return self._convert_id_to_token(ids)
        toks = []
        special_ids = set(self.all_special_ids) if skip_special_tokens else set()
        for i in ids:
            i = int(i)
            if i in special_ids:
                continue
            toks.append(self.convert_ids_to_tokens(i))
        return toks

    def encode(self, text, add_special_tokens=False):
        ids = self.convert_tokens_to_ids(self.tokenize(text))
        if add_special_tokens:
            ids = self.build_inputs_with_special_tokens(ids)
        return ids

    def build_inputs_with_special_tokens(self, token_ids_0, token_ids_1=None):
        return token_ids_0 if token_ids_1 is None else token_ids_0 + token_ids_1

    def decode(self, token_ids, skip_special_tokens=False):
        tokens = self.convert_ids_to_tokens(token_ids,
                                            skip_special_tokens=skip_special_tokens)
        return self.convert_tokens_to_string(tokens)

    def __call__(self, text, **kwargs):
        if isinstance(text, (list, tuple)):
            return {"input_ids": [self.encode(t, **kwargs) for t in text]}
        return {"input_ids": self.encode(text, **kwargs)}

    # -- persistence --------------------------------------------------------

    def save_pretrained(self, save_directory):
        os.makedirs(save_directory, exist_ok=True)
        special_map = {
            k: getattr(self, k)
            for k in self.SPECIAL_TOKENS_ATTRIBUTES
            if getattr(self, k, None)
        }
        with open(os.path.join(save_directory, SPECIAL_TOKENS_MAP_FILE), "w") as f:
            json.dump(special_map, f)
        if self.added_tokens_encoder:
            with open(os.path.join(save_directory, ADDED_TOKENS_FILE), "w") as f:
                json.dump(self.added_tokens_encoder, f)
        return self.save_vocabulary(save_directory)

    def save_vocabulary(self, save_directory):
        raise NotImplementedError

    @classmethod
    def from_pretrained(cls, directory, **kwargs):
        files = {
            name: os.path.join(directory, fname)
            for name, fname in cls.vocab_files_names.items()
        }
        sp_file = os.path.join(directory, SPECIAL_TOKENS_MAP_FILE)
        if os.path.exists(sp_file):
            with open(sp_file) as f:
                kwargs = {**json.load(f), **kwargs}
        tok = cls(**files, **kwargs)
        added_file = os.path.join(directory, ADDED_TOKENS_FILE)
        if os.path.exists(added_file):
            with open(added_file) as f:
                added = json.load(f)
            tok.added_tokens_encoder.update(added)
            tok.added_tokens_decoder.update({v: k for k, v in added.items()})
        return tok
