"""Mixup / CutMix batch augmentation for the CV path (reference: mixup hook in
libai/data/build.py:272-397 + engine/default.py:509-515)."""

import numpy as np
import torch

__all__ = ["Mixup"]


class Mixup:
    def __init__(self, mixup_alpha=0.8, cutmix_alpha=1.0, prob=1.0,
                 switch_prob=0.5, label_smoothing=0.1, num_classes=1000):
        self.mixup_alpha = mixup_alpha
        self.cutmix_alpha = cutmix_alpha
        self.prob = prob
        self.switch_prob = switch_prob
        self.label_smoothing = label_smoothing
        self.num_classes = num_classes

    def _one_hot(self, labels, lam, perm):
        off = self.label_smoothing / self.num_classes
        on = 1.0 - self.label_smoothing + off
        y = torch.full((labels.size(0), self.num_classes), off,
                       device=labels.device)
        y.scatter_(1, labels[:, None], on)
        return lam * y + (1 - lam) * y[perm]

    def __call__(self, images, labels):
        # All randomness comes from torch's generator: the trainer seeds
        # torch identically within each TP group (same_seed_for_tp_group),
        # so TP ranks apply the SAME mix to the replicated batch.  np.random
        # is seeded per-global-rank and would desync TP activations.
        if float(torch.rand(())) > self.prob:
            return images, self._one_hot(labels, 1.0, torch.arange(len(labels)))
        perm = torch.randperm(images.size(0), device=images.device)
        use_cutmix = float(torch.rand(())) < self.switch_prob and self.cutmix_alpha > 0
        if use_cutmix:
            lam = float(torch.distributions.Beta(
                self.cutmix_alpha, self.cutmix_alpha).sample())
            H, W = images.shape[-2:]
            rh, rw = int(H * np.sqrt(1 - lam)), int(W * np.sqrt(1 - lam))
            cy, cx = int(torch.randint(H, ())), int(torch.randint(W, ()))
            y1, y2 = max(cy - rh // 2, 0), min(cy + rh // 2, H)
            x1, x2 = max(cx - rw // 2, 0), min(cx + rw // 2, W)
            images[..., y1:y2, x1:x2] = images[perm][..., y1:y2, x1:x2]
            lam = 1.0 - (y2 - y1) * (x2 - x1) / (H * W)
        else:
            lam = float(torch.distributions.Beta(
                self.mixup_alpha, self.mixup_alpha).sample())
            images = lam * images + (1 - lam) * images[perm]
        return images, self._one_hot(labels, lam, perm)
