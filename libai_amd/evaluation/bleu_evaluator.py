"""Corpus BLEU evaluator (reference: libai/evaluation/bleu.py)."""

import math
from collections import Counter, OrderedDict

from .evaluator import DatasetEvaluator

__all__ = ["BleuEvaluator", "corpus_bleu"]


def _ngrams(tokens, n):
    return Counter(tuple(tokens[i : i + n]) for i in range(len(tokens) - n + 1))


def corpus_bleu(candidates, references, max_n=4):
    """candidates/references: lists of token lists (one reference each)."""
    p_num = [0] * max_n
    p_den = [0] * max_n
    cand_len = ref_len = 0
    for cand, ref in zip(candidates, references):
        cand_len += len(cand)
        ref_len += len(ref)
        for n in range(1, max_n + 1):
            cn = _ngrams(cand, n)
            rn = _ngrams(ref, n)
            p_num[n - 1] += sum((cn & rn).values())
            p_den[n - 1] += max(sum(cn.values()), 0)
    if min(p_den) == 0 or min(p_num) == 0:
        return 0.0
    log_p = sum(math.log(p_num[i] / p_den[i]) for i in range(max_n)) / max_n
    bp = 1.0 if cand_len > ref_len else math.exp(1 - ref_len / max(cand_len, 1))
    return bp * math.exp(log_p)


class BleuEvaluator(DatasetEvaluator):
    def __init__(self, tokenizer=None, max_n=4):
        self.tokenizer = tokenizer
        self.max_n = max_n
        self._cands, self._refs = [], []

    def reset(self):
        self._cands, self._refs = [], []

    def process(self, inputs, outputs):
        cands = outputs.get("sequences", outputs.get("prediction_ids"))
        refs = inputs.get("labels", inputs.get("lm_labels"))
        if cands is None or refs is None:
            return
        for c, r in zip(cands, refs):
            c = [int(x) for x in c.tolist()] if hasattr(c, "tolist") else list(c)
            r = [int(x) for x in r.tolist()] if hasattr(r, "tolist") else list(r)
            r = [x for x in r if x >= 0]
            self._cands.append(c)
            self._refs.append(r)

    def evaluate(self):
        if not self._cands:
            return {}
        return {"bleu": OrderedDict(
            bleu=100.0 * corpus_bleu(self._cands, self._refs, self.max_n)
        )}
