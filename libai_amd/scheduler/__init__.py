from .lr_scheduler import (
    WarmupConstantLR,
    WarmupCosineAnnealingLR,
    WarmupCosineLR,
    WarmupExponentialLR,
    WarmupMultiStepLR,
    WarmupPolynomialLR,
    WarmupStepLR,
)

__all__ = [
    "WarmupCosineLR",
    "WarmupCosineAnnealingLR",
    "WarmupStepLR",
    "WarmupMultiStepLR",
    "WarmupExponentialLR",
    "WarmupPolynomialLR",
    "WarmupConstantLR",
]
