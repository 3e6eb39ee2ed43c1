"""Model build helper (reference: libai/models/build.py:19-53)."""

from ..config import instantiate

__all__ = ["build_model"]


def build_model(cfg):
    return instantiate(cfg)
