"""Recursive materialization of ``_target_`` config nodes.

Reference behavior: libai/config/instantiate.py:130-201.
"""

import dataclasses
from collections import abc

from .lazy import ConfigDict, locate

__all__ = ["instantiate", "dump_dataclass"]


def dump_dataclass(obj):
    """Convert a dataclass (possibly nested) into a config tree with _target_."""
    assert dataclasses.is_dataclass(obj) and not isinstance(obj, type), (
        f"dump_dataclass expects a dataclass instance, got {obj!r}"
    )
    ret = ConfigDict()
    ret["_target_"] = type(obj).__module__ + "." + type(obj).__qualname__
    for f in dataclasses.fields(obj):
        v = getattr(obj, f.name)
        if dataclasses.is_dataclass(v) and not isinstance(v, type):
            v = dump_dataclass(v)
        if isinstance(v, (list, tuple)):
            v = [dump_dataclass(x) if dataclasses.is_dataclass(x) else x for x in v]
        ret[f.name] = v
    return ret


def instantiate(cfg):
    """Recursively build the object described by a config tree.

    * dict/ConfigDict with ``_target_`` -> call the target with instantiated kwargs
    * ``_recursive_=False`` on a node disables recursion below it
    * lists/tuples are instantiated element-wise
    * everything else returns as-is
    """
    if isinstance(cfg, (list, tuple)):
        return type(cfg)(instantiate(x) for x in cfg) if isinstance(cfg, tuple) else [
            instantiate(x) for x in cfg
        ]

    if isinstance(cfg, abc.Mapping) and "_target_" in cfg:
        recursive = cfg.get("_recursive_", True)
        kwargs = {}
        for k, v in cfg.items():
            if k in ("_target_", "_recursive_"):
                continue
            kwargs[k] = instantiate(v) if recursive else v
        target = cfg["_target_"]
        if isinstance(target, str):
            target = locate(target)
        if not callable(target):
            raise TypeError(f"_target_ {cfg['_target_']!r} is not callable")
        try:
            return target(**kwargs)
        except TypeError as e:
            raise TypeError(f"error instantiating {target!r}: {e}") from e

    if isinstance(cfg, abc.Mapping):
        # plain mapping: instantiate values but keep the mapping type
        return ConfigDict({k: instantiate(v) for k, v in cfg.items()})

    return cfg
