from .cls_evaluator import ClsEvaluator
from .evaluator import (
    DatasetEvaluator,
    DatasetEvaluators,
    flatten_results_dict,
    inference_on_dataset,
)
from .ppl_evaluator import PPLEvaluator

__all__ = [
    "DatasetEvaluator",
    "DatasetEvaluators",
    "inference_on_dataset",
    "flatten_results_dict",
    "ClsEvaluator",
    "PPLEvaluator",
]
