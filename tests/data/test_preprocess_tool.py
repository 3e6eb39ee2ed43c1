"""tools/preprocess_data.py end-to-end: jsonl -> .bin/.idx -> GPT2Dataset."""

import json
import subprocess
import sys
import os

import pytest


def test_preprocess_jsonl_roundtrip(tmp_path):
    root = os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    # word-level BERT vocab (WordPiece tokenizer needs no merges file)
    vocab = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]", "the", "quick",
             "brown", "fox", "dog", "runs", "over", "lazy", "."]
    vf = tmp_path / "vocab.txt"
    vf.write_text("\n".join(vocab) + "\n")
    src = tmp_path / "corpus.jsonl"
    with open(src, "w") as f:
        for _ in range(20):
            f.write(json.dumps({"text": "the quick brown fox runs over the lazy dog ."}) + "\n")
    prefix = str(tmp_path / "out")
    r = subprocess.run(
        [sys.executable, os.path.join(root, "tools", "preprocess_data.py"),
         "--input", str(src), "--tokenizer-type", "BertTokenizer",
         "--vocab-file", str(vf), "--output-prefix", prefix],
        capture_output=True, text=True, cwd=root,
    )
    assert r.returncode == 0, r.stderr
    assert os.path.exists(prefix + ".bin") and os.path.exists(prefix + ".idx")

    from libai_amd.data.indexed_dataset import MMapIndexedDataset

    ds = MMapIndexedDataset(prefix)
    assert len(ds) == 20
    ids = list(ds[0])
    assert len(ids) == 10  # 10 word-level tokens per line
