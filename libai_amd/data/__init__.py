from .build import (
    build_image_test_loader,
    build_image_train_loader,
    build_nlp_test_loader,
    build_nlp_train_loader,
    build_train_valid_test_loaders,
    trivial_batch_collator,
)
from .samplers import CyclicSampler, SingleRoundSampler
from .structures import DistTensorData, Instance

__all__ = [
    "DistTensorData",
    "Instance",
    "CyclicSampler",
    "SingleRoundSampler",
    "trivial_batch_collator",
    "build_nlp_train_loader",
    "build_nlp_test_loader",
    "build_image_train_loader",
    "build_image_test_loader",
    "build_train_valid_test_loaders",
]
