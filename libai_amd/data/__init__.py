from .blendable import BlendableDataset
from .build import (
    build_image_test_loader,
    build_image_train_loader,
    build_nlp_test_loader,
    build_nlp_train_loader,
    build_nlp_train_val_test_loader,
    build_train_valid_test_loaders,
    trivial_batch_collator,
)
from .mixup import Mixup
from .samplers import CyclicSampler, SingleRoundSampler
from .structures import DistTensorData, Instance

__all__ = [
    "DistTensorData",
    "Instance",
    "BlendableDataset",
    "build_nlp_train_val_test_loader",
    "CyclicSampler",
    "Mixup",
    "SingleRoundSampler",
    "trivial_batch_collator",
    "build_nlp_train_loader",
    "build_nlp_test_loader",
    "build_image_train_loader",
    "build_image_test_loader",
    "build_train_valid_test_loaders",
]
