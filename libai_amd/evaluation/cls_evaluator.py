"""Top-k classification evaluator (reference: libai/evaluation/cls_evaluator.py)."""

from collections import OrderedDict

import torch

from .evaluator import DatasetEvaluator

__all__ = ["ClsEvaluator"]


class ClsEvaluator(DatasetEvaluator):
    def __init__(self, topk=(1, 5)):
        self.topk = topk
        self._predictions = []

    def reset(self):
        self._predictions = []

    def process(self, inputs, outputs):
        logits = outputs.get("prediction_scores")
        labels = inputs.get("labels")
        if logits is None or labels is None:
            return
        maxk = min(max(self.topk), logits.shape[-1])
        pred = logits.float().topk(maxk, dim=-1).indices  # [N, maxk]
        correct = pred.eq(labels.view(-1, 1))
        entry = {f"top{k}_num": int(correct[:, :k].any(dim=1).sum()) for k in self.topk}
        entry["num_samples"] = labels.numel()
        self._predictions.append(entry)

    def evaluate(self):
        if not self._predictions:
            return {}
        total = sum(p["num_samples"] for p in self._predictions)
        results = OrderedDict()
        for k in self.topk:
            hit = sum(p[f"top{k}_num"] for p in self._predictions)
            results[f"Acc@{k}"] = 100.0 * hit / max(total, 1)
        return {"cls": results}
