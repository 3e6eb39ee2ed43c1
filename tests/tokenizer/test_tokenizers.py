import json
import os
import tempfile

import pytest

from libai_amd.tokenizer import BertTokenizer, GPT2Tokenizer


@pytest.fixture()
def bert_vocab(tmp_path):
    vocab = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]", "the", "quick", "brown",
             "fox", "jump", "##ed", "##s", "over", "lazy", "dog", ",", "."]
    p = tmp_path / "vocab.txt"
    p.write_text("\n".join(vocab) + "\n")
    return str(p)


def test_bert_wordpiece(bert_vocab):
    tok = BertTokenizer(bert_vocab)
    toks = tok.tokenize("The quick brown fox jumped.")
    assert toks == ["the", "quick", "brown", "fox", "jump", "##ed", "."]
    ids = tok.encode("the quick", add_special_tokens=True)
    assert ids[0] == tok.vocab["[CLS]"] and ids[-1] == tok.vocab["[SEP]"]
    assert tok.decode(tok.encode("the quick brown")) == "the quick brown"
    assert tok.tokenize("unknownword") == ["[UNK]"]


def test_bert_padded_vocab(bert_vocab):
    tok = BertTokenizer(bert_vocab)
    assert tok.padded_vocab_size(8) % 8 == 0
    assert tok.padded_vocab_size(8) >= tok.vocab_size


@pytest.fixture()
def gpt2_files(tmp_path):
    # tiny byte-BPE: enough to merge "he" and "the"-ish pieces
    from libai_amd.tokenizer.gpt2_tokenizer import bytes_to_unicode

    b2u = bytes_to_unicode()
    base_tokens = [b2u[i] for i in range(256)]
    merges = [("t", "h"), ("th", "e"), ("Ġ", "t"), ("Ġt", "he")]
    merged_tokens = ["th", "the", "Ġt", "Ġthe"]
    vocab = {t: i for i, t in enumerate(base_tokens + merged_tokens + ["<|endoftext|>"])}
    vp = tmp_path / "vocab.json"
    vp.write_text(json.dumps(vocab))
    mp = tmp_path / "merges.txt"
    mp.write_text("#version: 0.2\n" + "\n".join(" ".join(m) for m in merges) + "\n")
    return str(vp), str(mp)


def test_gpt2_bpe_roundtrip(gpt2_files):
    vocab_file, merges_file = gpt2_files
    tok = GPT2Tokenizer(vocab_file, merges_file)
    text = "the theory"
    ids = tok.encode(text)
    assert tok.decode(ids) == text
    # merges applied: "the" should be a single token at the start
    assert tok.tokenize("the")[0] == "the"


def test_gpt2_handles_arbitrary_bytes(gpt2_files):
    vocab_file, merges_file = gpt2_files
    tok = GPT2Tokenizer(vocab_file, merges_file)
    text = "café 中文!"
    assert tok.decode(tok.encode(text)) == text


def test_save_load_roundtrip(gpt2_files, tmp_path):
    vocab_file, merges_file = gpt2_files
    tok = GPT2Tokenizer(vocab_file, merges_file)
    d = str(tmp_path / "saved")
    tok.save_pretrained(d)
    tok2 = GPT2Tokenizer.from_pretrained(d)
    assert tok2.encode("the theory") == tok.encode("the theory")
