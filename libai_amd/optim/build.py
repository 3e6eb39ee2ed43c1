"""Optimizer param-group construction (reference: libai/optim/build.py:42-162).

Per-param overrides: norm-layer weights and biases get weight_decay 0 by
default; clip_grad plumbs into FusedAdamW (not into param groups — the HIP
update applies the clip scale in the same pass).
"""

import copy
import itertools

import torch
from torch import nn

from ..config import instantiate, try_get_key

__all__ = ["build_optimizer", "get_default_optimizer_params"]


def build_optimizer(cfg, model):
    """cfg is a LazyCall of an optimizer whose params is a LazyCall of
    get_default_optimizer_params (reference configs/common/optim.py shape)."""
    cfg = copy.deepcopy(cfg)
    if "params" in cfg and isinstance(cfg["params"], dict):
        cfg["params"]["model"] = model
    return instantiate(cfg)


def get_default_optimizer_params(
    model,
    base_lr=None,
    weight_decay=None,
    weight_decay_norm=0.0,
    weight_decay_bias=0.0,
    clip_grad_max_norm=None,
    clip_grad_norm_type=None,
    overrides=None,
):
    """Per-parameter groups with norm/bias weight-decay overrides."""
    if overrides is None:
        overrides = {}
    defaults = {}
    if base_lr is not None:
        defaults["lr"] = base_lr
    if weight_decay is not None:
        defaults["weight_decay"] = weight_decay

    norm_module_types = (
        nn.BatchNorm1d, nn.BatchNorm2d, nn.BatchNorm3d, nn.GroupNorm,
        nn.InstanceNorm1d, nn.InstanceNorm2d, nn.InstanceNorm3d, nn.LayerNorm,
        nn.LocalResponseNorm,
    )
    try:
        from ..layers.layer_norm import LayerNorm as LibaiLN
        from ..layers.layer_norm import RMSLayerNorm as LibaiRMS

        norm_module_types = norm_module_types + (LibaiLN, LibaiRMS)
    except ImportError:
        pass

    params = []
    memo = set()
    for module in model.modules():
        for name, value in module.named_parameters(recurse=False):
            if not value.requires_grad or value in memo:
                continue
            memo.add(value)
            hyperparams = copy.copy(defaults)
            if isinstance(module, norm_module_types) and weight_decay_norm is not None:
                hyperparams["weight_decay"] = weight_decay_norm
            elif name == "bias" and weight_decay_bias is not None:
                hyperparams["weight_decay"] = weight_decay_bias
            hyperparams.update(overrides.get(name, {}))
            params.append({"params": [value], **hyperparams})
    return reduce_param_groups(params)


def _expand_param_groups(params):
    ret = {}
    for item in params:
        assert "params" in item
        cur = {k: v for k, v in item.items() if k != "params"}
        for p in item["params"]:
            ret.setdefault(id(p), {"params": [p]})
            ret[id(p)].update(cur)
    return list(ret.values())


def reduce_param_groups(params):
    """Merge groups with identical hyperparameters (reference: build.py:129-162)."""
    params = _expand_param_groups(params)
    groups = {}
    for item in params:
        key = tuple((k, v) for k, v in sorted(item.items()) if k != "params")
        groups.setdefault(key, []).extend(item["params"])
    ret = []
    for key, plist in groups.items():
        cur = dict(key)
        cur["params"] = plist
        ret.append(cur)
    return ret
