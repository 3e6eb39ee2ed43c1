"""Perplexity evaluator (reference: libai/evaluation/ppl_evaluator.py)."""

import math
from collections import OrderedDict

from .evaluator import DatasetEvaluator

__all__ = ["PPLEvaluator"]


class PPLEvaluator(DatasetEvaluator):
    def __init__(self):
        self._sum_nll = 0.0
        self._num_tokens = 0

    def reset(self):
        self._sum_nll = 0.0
        self._num_tokens = 0

    def process(self, inputs, outputs):
        # models emit per-batch mean loss under key *loss*
        for k, v in outputs.items():
            if "loss" in k:
                n = inputs.get("labels", inputs.get("lm_labels"))
                ntok = n.numel() if n is not None else 1
                self._sum_nll += float(v) * ntok
                self._num_tokens += ntok

    def evaluate(self):
        if self._num_tokens == 0:
            return {}
        nll = self._sum_nll / self._num_tokens
        return {"ppl": OrderedDict(ppl=math.exp(min(nll, 50.0)), nll=nll)}
