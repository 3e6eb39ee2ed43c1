import os
import tempfile

import pytest

from libai_amd.config import (
    ConfigDict,
    LazyCall,
    LazyConfig,
    configurable,
    instantiate,
    try_get_key,
)


class Model:
    def __init__(self, a, b=2, sub=None):
        self.a, self.b, self.sub = a, b, sub


def test_lazycall_builds_config_node():
    node = LazyCall(Model)(a=1, b=3)
    assert node["_target_"] is Model
    assert node.a == 1 and node.b == 3


def test_instantiate_recursive():
    node = LazyCall(Model)(a=1, sub=LazyCall(Model)(a=5))
    obj = instantiate(node)
    assert isinstance(obj, Model) and isinstance(obj.sub, Model)
    assert obj.sub.a == 5


def test_instantiate_string_target():
    node = ConfigDict({"_target_": "collections.OrderedDict"})
    obj = instantiate(node)
    from collections import OrderedDict

    assert isinstance(obj, OrderedDict)


def test_config_load_and_relative_import():
    with tempfile.TemporaryDirectory() as d:
        os.makedirs(os.path.join(d, "common"))
        with open(os.path.join(d, "common", "base.py"), "w") as f:
            f.write("train = dict(lr=0.1, iters=100)\n")
        with open(os.path.join(d, "main.py"), "w") as f:
            f.write(
                "from .common.base import train\n"
                "train['lr'] = 0.5\n"
                "model = dict(width=3)\n"
            )
        cfg = LazyConfig.load(os.path.join(d, "main.py"))
        assert cfg.train.lr == 0.5
        assert cfg.train.iters == 100
        assert cfg.model.width == 3


def test_apply_overrides():
    cfg = ConfigDict({"train": {"lr": 0.1, "nested": {"x": 1}}})
    LazyConfig.apply_overrides(
        cfg, ["train.lr=0.5", "train.nested.x=7", "train.name=adam", "train.flag=True"]
    )
    assert cfg.train.lr == 0.5
    assert cfg.train.nested.x == 7
    assert cfg.train.name == "adam"
    assert cfg.train.flag is True


def test_save_yaml_roundtrippable():
    cfg = ConfigDict({"train": {"lr": 0.1}, "model": LazyCall(Model)(a=1)})
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "cfg.yaml")
        LazyConfig.save(cfg, path)
        assert os.path.exists(path)


def test_try_get_key():
    cfg = ConfigDict({"a": {"b": {"c": 3}}})
    assert try_get_key(cfg, "a.b.c") == 3
    assert try_get_key(cfg, "a.b.missing", "x.y", default=9) == 9


class Configurable:
    @configurable
    def __init__(self, a, b=2):
        self.a, self.b = a, b

    @classmethod
    def from_config(cls, cfg):
        return {"a": cfg.A, "b": cfg.B}


def test_configurable_both_paths():
    direct = Configurable(1, b=5)
    assert (direct.a, direct.b) == (1, 5)
    from_cfg = Configurable(ConfigDict({"A": 7, "B": 8}))
    assert (from_cfg.a, from_cfg.b) == (7, 8)


def test_config_deepcopy_independent():
    import copy

    cfg = ConfigDict({"train": {"lr": 0.1}})
    cfg2 = copy.deepcopy(cfg)
    cfg2.train.lr = 0.9
    assert cfg.train.lr == 0.1
