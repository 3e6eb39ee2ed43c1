#!/usr/bin/env python3
"""jsonl -> .bin/.idx indexed dataset (reference: tools/preprocess_data.py).

Usage:
  python tools/preprocess_data.py --input corpus.jsonl --json-key text \
      --tokenizer-type GPT2Tokenizer --vocab-file vocab.json \
      --merges-file merges.txt --output-prefix my_corpus --append-eod
"""

import argparse
import json
import multiprocessing
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import numpy as np

from libai_amd.data.indexed_dataset import (
    MMapIndexedDatasetBuilder,
    best_fitting_dtype,
    data_file_path,
    index_file_path,
)

_TOK = None


def _init_tokenizer(args):
    global _TOK
    from libai_amd import tokenizer as tok_mod

    if args.tokenizer_type == "GPT2Tokenizer":
        _TOK = tok_mod.GPT2Tokenizer(args.vocab_file, args.merges_file)
    elif args.tokenizer_type == "BertTokenizer":
        _TOK = tok_mod.BertTokenizer(args.vocab_file)
    elif args.tokenizer_type == "T5Tokenizer":
        _TOK = tok_mod.T5Tokenizer(args.vocab_file)
    else:
        raise ValueError(f"unknown tokenizer {args.tokenizer_type}")
    return _TOK


def _encode(line_args):
    line, key, append_eod = line_args
    try:
        text = json.loads(line)[key]
    except (json.JSONDecodeError, KeyError):
        return None
    ids = _TOK.encode(text)
    if append_eod and _TOK.eos_token:
        ids.append(_TOK.convert_tokens_to_ids(_TOK.eos_token))
    return ids


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--input", required=True)
    p.add_argument("--json-key", default="text")
    p.add_argument("--tokenizer-type", default="GPT2Tokenizer")
    p.add_argument("--vocab-file")
    p.add_argument("--merges-file")
    p.add_argument("--output-prefix", required=True)
    p.add_argument("--append-eod", action="store_true")
    p.add_argument("--workers", type=int, default=1)
    args = p.parse_args()

    tok = _init_tokenizer(args)
    dtype = best_fitting_dtype(len(tok))
    builder = MMapIndexedDatasetBuilder(data_file_path(args.output_prefix), dtype)

    t0 = time.time()
    n_docs = n_tokens = 0
    with open(args.input) as f:
        if args.workers > 1:
            with multiprocessing.Pool(
                args.workers, initializer=_init_tokenizer, initargs=(args,)
            ) as pool:
                for ids in pool.imap(
                    _encode, ((l, args.json_key, args.append_eod) for l in f),
                    chunksize=32,
                ):
                    if not ids:
                        continue
                    builder.add_item(np.array(ids))
                    builder.end_document()
                    n_docs += 1
                    n_tokens += len(ids)
        else:
            for line in f:
                ids = _encode((line, args.json_key, args.append_eod))
                if not ids:
                    continue
                builder.add_item(np.array(ids))
                builder.end_document()
                n_docs += 1
                n_tokens += len(ids)
    builder.finalize(index_file_path(args.output_prefix))
    print(
        f"wrote {n_docs} docs / {n_tokens} tokens to {args.output_prefix}.bin/.idx "
        f"({time.time() - t0:.1f}s)"
    )


if __name__ == "__main__":
    main()
