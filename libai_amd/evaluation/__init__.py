from .bleu_evaluator import BleuEvaluator, corpus_bleu
from .cls_evaluator import ClsEvaluator
from .evaluator import (
    DatasetEvaluator,
    DatasetEvaluators,
    flatten_results_dict,
    inference_on_dataset,
)
from .ppl_evaluator import PPLEvaluator
from .reg_evaluator import RegEvaluator

__all__ = [
    "DatasetEvaluator",
    "DatasetEvaluators",
    "inference_on_dataset",
    "flatten_results_dict",
    "ClsEvaluator",
    "PPLEvaluator",
    "BleuEvaluator",
    "corpus_bleu",
    "RegEvaluator",
]
