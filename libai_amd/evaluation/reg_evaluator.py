"""Regression evaluator: MSE / MAE / pearson (reference: libai/evaluation/reg)."""

import math
from collections import OrderedDict

import torch

from .evaluator import DatasetEvaluator

__all__ = ["RegEvaluator"]


class RegEvaluator(DatasetEvaluator):
    def __init__(self):
        self._preds, self._labels = [], []

    def reset(self):
        self._preds, self._labels = [], []

    def process(self, inputs, outputs):
        p = outputs.get("prediction_scores")
        l = inputs.get("labels")
        if p is None or l is None:
            return
        self._preds.append(p.detach().float().reshape(-1).cpu())
        self._labels.append(l.detach().float().reshape(-1).cpu())

    def evaluate(self):
        if not self._preds:
            return {}
        p = torch.cat(self._preds)
        l = torch.cat(self._labels)
        mse = float(((p - l) ** 2).mean())
        mae = float((p - l).abs().mean())
        pc = float(
            ((p - p.mean()) * (l - l.mean())).sum()
            / (p.std(unbiased=False) * l.std(unbiased=False) * len(p) + 1e-12)
        )
        return {"regression": OrderedDict(mse=mse, mae=mae, pearson=pc)}
