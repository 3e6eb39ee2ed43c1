"""Loader for the in-tree gfx950 HIP extension (libai_amd/_C.so).

Policy: on a GPU box the extension is REQUIRED — ops raise rather than fall
back to eager PyTorch, so a silently-slow path can never masquerade as the
native one.  On CPU-only hosts ops use their PyTorch reference
implementations (which are also the numerics oracle for the GPU tests).
"""

import os

import torch

_EXT = None
_TRIED = False


def _load():
    global _EXT, _TRIED
    if _TRIED:
        return _EXT
    _TRIED = True
    so_path = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "_C.so")
    try:
        from libai_amd import _C  # noqa: F401

        _EXT = _C
    except ImportError as e:
        if os.path.exists(so_path):
            raise ImportError(f"libai_amd/_C.so exists but failed to import: {e}") from e
        _EXT = None
    return _EXT


def has_ext():
    return _load() is not None


def ext():
    mod = _load()
    if mod is None:
        raise RuntimeError(
            "libai_amd HIP extension not built. Run `python libai_amd/csrc/build.py` "
            "(gfx950 cross-compile works without a GPU). GPU ops refuse to fall back "
            "to eager PyTorch by design."
        )
    return mod


def use_hip(x: torch.Tensor) -> bool:
    """HIP kernels serve CUDA(ROCm) tensors; CPU tensors use reference impls."""
    return x.is_cuda


_seed_gen = None


def draw_seed() -> int:
    """Philox seed for recompute-in-backward dropout masks.

    Drawn from the torch CPU generator so torch.utils.checkpoint's RNG-state
    preservation makes activation-checkpoint recompute reproduce the same
    masks.
    """
    return int(torch.randint(0, 2**62, (1,)).item())
