from .bert_dataset import BertDataset, create_masked_lm_predictions
from .gpt_dataset import GPT2Dataset
from .synthetic import SyntheticBertDataset, SyntheticGPTDataset, SyntheticImageDataset
from .t5_dataset import T5Dataset

__all__ = [
    "GPT2Dataset",
    "BertDataset",
    "T5Dataset",
    "create_masked_lm_predictions",
    "SyntheticGPTDataset",
    "SyntheticBertDataset",
    "SyntheticImageDataset",
]
