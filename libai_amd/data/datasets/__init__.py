from .synthetic import SyntheticBertDataset, SyntheticGPTDataset, SyntheticImageDataset

__all__ = ["SyntheticGPTDataset", "SyntheticBertDataset", "SyntheticImageDataset"]
