"""LR schedulers with warmup (reference: libai/scheduler/lr_scheduler.py:23-257).

All factories return torch LambdaLR schedulers composing a warmup phase
(linear or constant) with the base decay, matching the reference's
WarmUp{Cosine,CosineAnnealing,Step,MultiStep,Exponential,Polynomial}LR set.
Schedulers step once per ITERATION.
"""

import math

from torch.optim.lr_scheduler import LambdaLR

__all__ = [
    "WarmupCosineLR",
    "WarmupCosineAnnealingLR",
    "WarmupStepLR",
    "WarmupMultiStepLR",
    "WarmupExponentialLR",
    "WarmupPolynomialLR",
    "WarmupConstantLR",
]


def _warmup_factor(method, it, warmup_iter, warmup_factor):
    if warmup_iter <= 0 or it >= warmup_iter:
        return 1.0
    if method == "constant":
        return warmup_factor
    if method == "linear":
        alpha = it / warmup_iter
        return warmup_factor * (1 - alpha) + alpha
    raise ValueError(f"unknown warmup method {method!r}")


def WarmupCosineLR(optimizer, max_iter, warmup_iter=0, warmup_factor=0.0001,
                   alpha=0.0, warmup_method="linear", **kwargs):
    """Cosine from base_lr to alpha*base_lr over max_iter, with warmup."""

    def fn(it):
        w = _warmup_factor(warmup_method, it, warmup_iter, warmup_factor)
        if it < warmup_iter:
            return w
        progress = min(1.0, (it - warmup_iter) / max(1, max_iter - warmup_iter))
        cos = 0.5 * (1 + math.cos(math.pi * progress))
        return alpha + (1 - alpha) * cos

    return LambdaLR(optimizer, fn)


WarmupCosineAnnealingLR = WarmupCosineLR


def WarmupStepLR(optimizer, step_size, gamma=0.1, warmup_iter=0,
                 warmup_factor=0.0001, warmup_method="linear", **kwargs):
    def fn(it):
        w = _warmup_factor(warmup_method, it, warmup_iter, warmup_factor)
        if it < warmup_iter:
            return w
        return gamma ** ((it - warmup_iter) // step_size)

    return LambdaLR(optimizer, fn)


def WarmupMultiStepLR(optimizer, milestones, gamma=0.1, warmup_iter=0,
                      warmup_factor=0.0001, warmup_method="linear", **kwargs):
    milestones = sorted(milestones)

    def fn(it):
        w = _warmup_factor(warmup_method, it, warmup_iter, warmup_factor)
        if it < warmup_iter:
            return w
        return gamma ** sum(1 for m in milestones if it >= m)

    return LambdaLR(optimizer, fn)


def WarmupExponentialLR(optimizer, gamma, warmup_iter=0, warmup_factor=0.0001,
                        warmup_method="linear", **kwargs):
    def fn(it):
        w = _warmup_factor(warmup_method, it, warmup_iter, warmup_factor)
        if it < warmup_iter:
            return w
        return gamma ** (it - warmup_iter)

    return LambdaLR(optimizer, fn)


def WarmupPolynomialLR(optimizer, max_iter, end_learning_rate_ratio=0.0, power=1.0,
                       warmup_iter=0, warmup_factor=0.0001, warmup_method="linear",
                       cycle=False, **kwargs):
    def fn(it):
        w = _warmup_factor(warmup_method, it, warmup_iter, warmup_factor)
        if it < warmup_iter:
            return w
        decay_iter = max(1, max_iter - warmup_iter)
        x = it - warmup_iter
        if cycle and x > 0:
            decay_iter = decay_iter * math.ceil(x / decay_iter)
        x = min(x, decay_iter)
        return (1 - end_learning_rate_ratio) * (1 - x / decay_iter) ** power + (
            end_learning_rate_ratio
        )

    return LambdaLR(optimizer, fn)


def WarmupConstantLR(optimizer, warmup_iter=0, warmup_factor=0.0001,
                     warmup_method="linear", **kwargs):
    def fn(it):
        return _warmup_factor(warmup_method, it, warmup_iter, warmup_factor)

    return LambdaLR(optimizer, fn)
