"""build_tokenizer with TP-divisible vocab padding + eod aliasing
(reference: libai/tokenizer/build.py:23-33)."""

import logging

from ..config import instantiate, try_get_key
from ..utils import distributed as du

__all__ = ["build_tokenizer"]


def build_tokenizer(cfg):
    tokenizer = instantiate(cfg.tokenization.tokenizer)
    multiple = try_get_key(cfg, "tokenization.make_vocab_size_divisible_by", default=1)
    tp = du.get_dist_util().tensor_parallel_size
    padded = tokenizer.padded_vocab_size(multiple * tp)
    logging.getLogger(__name__).info(
        f"tokenizer vocab {len(tokenizer)} -> padded {padded} (x{multiple * tp})"
    )
    tokenizer.padded_vocab = padded
    # eod aliasing: GPT datasets reference tokenizer.eod
    append_eod = try_get_key(cfg, "tokenization.append_eod", default=False)
    if not hasattr(tokenizer, "eod"):
        eod_tok = tokenizer.eos_token or tokenizer.sep_token or tokenizer.pad_token
        if eod_tok is not None:
            tokenizer.eod = tokenizer.convert_tokens_to_ids(eod_tok)
    tokenizer.append_eod = append_eod
    return tokenizer
