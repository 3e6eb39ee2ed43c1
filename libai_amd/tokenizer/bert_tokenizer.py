"""BERT WordPiece tokenizer (reference: libai/tokenizer/tokenization_bert.py)."""

import collections
import os
import unicodedata

from .tokenization_base import PreTrainedTokenizer

__all__ = ["BertTokenizer", "BasicTokenizer", "WordpieceTokenizer"]


def load_vocab(vocab_file):
    vocab = collections.OrderedDict()
    with open(vocab_file, encoding="utf-8") as f:
        for idx, line in enumerate(f):
            vocab[line.rstrip("\n")] = idx
    return vocab


def _is_whitespace(ch):
    return ch in " \t\n\r" or unicodedata.category(ch) == "Zs"


def _is_control(ch):
    if ch in ("\t", "\n", "\r"):
        return False
    return unicodedata.category(ch).startswith("C")


def _is_punctuation(ch):
    cp = ord(ch)
    if (33 <= cp <= 47) or (58 <= cp <= 64) or (91 <= cp <= 96) or (123 <= cp <= 126):
        return True
    return unicodedata.category(ch).startswith("P")


class BasicTokenizer:
    def __init__(self, do_lower_case=True, never_split=None):
        self.do_lower_case = do_lower_case
        self.never_split = set(never_split or [])

    def tokenize(self, text, never_split=None):
        never = self.never_split | set(never_split or [])
        text = self._clean_text(text)
        text = self._tokenize_chinese_chars(text)
        tokens = text.strip().split()
        out = []
        for tok in tokens:
            if tok in never:
                out.append(tok)
                continue
            if self.do_lower_case:
                tok = tok.lower()
                tok = self._strip_accents(tok)
            out.extend(self._split_on_punc(tok, never))
        return " ".join(out).strip().split()

    def _clean_text(self, text):
        return "".join(
            " " if _is_whitespace(c) else c
            for c in text
            if ord(c) != 0 and ord(c) != 0xFFFD and not _is_control(c)
        )

    def _strip_accents(self, text):
        return "".join(
            c for c in unicodedata.normalize("NFD", text)
            if unicodedata.category(c) != "Mn"
        )

    def _split_on_punc(self, text, never):
        if text in never:
            return [text]
        out, cur = [], []
        for c in text:
            if _is_punctuation(c):
                if cur:
                    out.append("".join(cur))
                    cur = []
                out.append(c)
            else:
                cur.append(c)
        if cur:
            out.append("".join(cur))
        return out

    def _tokenize_chinese_chars(self, text):
        out = []
        for c in text:
            cp = ord(c)
            if self._is_chinese_char(cp):
                out.extend([" ", c, " "])
            else:
                out.append(c)
        return "".join(out)

    @staticmethod
    def _is_chinese_char(cp):
        return (
            0x4E00 <= cp <= 0x9FFF or 0x3400 <= cp <= 0x4DBF
            or 0x20000 <= cp <= 0x2A6DF or 0x2A700 <= cp <= 0x2B73F
            or 0x2B740 <= cp <= 0x2B81F or 0x2B820 <= cp <= 0x2CEAF
            or 0xF900 <= cp <= 0xFAFF or 0x2F800 <= cp <= 0x2FA1F
        )


class WordpieceTokenizer:
    def __init__(self, vocab, unk_token="[UNK]", max_input_chars_per_word=100):
        self.vocab = vocab
        self.unk_token = unk_token
        self.max_input_chars_per_word = max_input_chars_per_word

    def tokenize(self, text):
        output = []
        for token in text.strip().split():
            chars = list(token)
            if len(chars) > self.max_input_chars_per_word:
                output.append(self.unk_token)
                continue
            start, sub_tokens, bad = 0, [], False
            while start < len(chars):
                end = len(chars)
                cur = None
                while start < end:
                    substr = "".join(chars[start:end])
                    if start > 0:
                        substr = "##" + substr
                    if substr in self.vocab:
                        cur = substr
                        break
                    end -= 1
                if cur is None:
                    bad = True
                    break
                sub_tokens.append(cur)
                start = end
            output.extend([self.unk_token] if bad else sub_tokens)
        return output


class BertTokenizer(PreTrainedTokenizer):
    vocab_files_names = {"vocab_file": "vocab.txt"}

    def __init__(self, vocab_file, do_lower_case=True, do_basic_tokenize=True,
                 never_split=None, unk_token="[UNK]", sep_token="[SEP]",
                 pad_token="[PAD]", cls_token="[CLS]", mask_token="[MASK]", **kwargs):
        super().__init__(unk_token=unk_token, sep_token=sep_token, pad_token=pad_token,
                         cls_token=cls_token, mask_token=mask_token, **kwargs)
        self.vocab = load_vocab(vocab_file)
        self.ids_to_tokens = {v: k for k, v in self.vocab.items()}
        self.do_basic_tokenize = do_basic_tokenize
        if do_basic_tokenize:
            self.basic_tokenizer = BasicTokenizer(do_lower_case, never_split)
        self.wordpiece_tokenizer = WordpieceTokenizer(self.vocab, unk_token)

    @property
    def vocab_size(self):
        return len(self.vocab)

    def get_vocab(self):
        return dict(self.vocab)

    def _tokenize(self, text):
        if self.do_basic_tokenize:
            tokens = []
            for tok in self.basic_tokenizer.tokenize(text, self.all_special_tokens):
                tokens.extend(self.wordpiece_tokenizer.tokenize(tok))
            return tokens
        return self.wordpiece_tokenizer.tokenize(text)

    def _convert_token_to_id(self, token):
        return self.vocab.get(token, self.vocab.get(self.unk_token))

    def _convert_id_to_token(self, index):
        return self.ids_to_tokens.get(index, self.unk_token)

    def convert_tokens_to_string(self, tokens):
        return " ".join(tokens).replace(" ##", "").strip()

    def build_inputs_with_special_tokens(self, token_ids_0, token_ids_1=None):
        cls = [self._convert_token_to_id(self.cls_token)]
        sep = [self._convert_token_to_id(self.sep_token)]
        if token_ids_1 is None:
            return cls + token_ids_0 + sep
        return cls + token_ids_0 + sep + token_ids_1 + sep

    def save_vocabulary(self, save_directory):
        path = os.path.join(save_directory, "vocab.txt")
        with open(path, "w", encoding="utf-8") as f:
            for token, _ in sorted(self.vocab.items(), key=lambda kv: kv[1]):
                f.write(token + "\n")
        return (path,)
