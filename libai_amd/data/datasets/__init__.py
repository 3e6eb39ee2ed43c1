from .bert_dataset import BertDataset, create_masked_lm_predictions
from .cv_datasets import CIFAR10Dataset, ImageFolderDataset, MNISTDataset
from .gpt_dataset import GPT2Dataset
from .roberta_dataset import RobertaDataset
from .synthetic import SyntheticBertDataset, SyntheticGPTDataset, SyntheticImageDataset
from .t5_dataset import T5Dataset

__all__ = [
    "GPT2Dataset",
    "BertDataset",
    "T5Dataset",
    "RobertaDataset",
    "create_masked_lm_predictions",
    "CIFAR10Dataset",
    "MNISTDataset",
    "ImageFolderDataset",
    "SyntheticGPTDataset",
    "SyntheticBertDataset",
    "SyntheticImageDataset",
]
