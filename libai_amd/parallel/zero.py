"""ZeRO configuration helper (reference: cfg.train.zero_optimization,
configs/common/train.py:61-67 / graph_base.py:69-70).

The sharding itself lives in FusedAdamW (flat buckets make the DP slices
contiguous): stage 1 shards optimizer state, stage 2 additionally
reduce-scatters gradients so each rank only materializes its own grad slice
reduction.  This module just maps config -> optimizer kwargs.
"""

from ..config import try_get_key

__all__ = ["zero_stage_from_config"]


def zero_stage_from_config(cfg):
    enabled = try_get_key(cfg, "train.zero_optimization.enabled", default=False)
    if not enabled:
        return 0
    stage = int(try_get_key(cfg, "train.zero_optimization.stage", default=1))
    if stage > 2:
        raise NotImplementedError(
            "ZeRO stage 3 (parameter sharding) is not implemented yet; use 1 or 2"
        )
    return stage
