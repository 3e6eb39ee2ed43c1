"""Lazy configuration system.

A from-scratch, dependency-free re-implementation of the LazyCall/LazyConfig
API that the reference exposes (reference: libai/config/lazy.py:89-463).
Unlike the reference we do not depend on omegaconf/hydra: configs are plain
``ConfigDict`` trees (attribute-accessible dicts) built by executing Python
config files, and CLI overrides use a small dotted ``key=value`` grammar.

Semantics preserved from the reference:
  * ``LazyCall(cls)(**kw)`` produces a config node with ``_target_`` that is
    materialized later by :func:`libai_amd.config.instantiate`.
  * ``LazyConfig.load(path)`` executes the config file as Python with relative
    imports between config files patched to work from any cwd.
  * ``LazyConfig.apply_overrides(cfg, ["a.b.c=3", ...])`` applies dotted
    overrides with Python-literal values.
  * ``LazyConfig.save`` round-trips to YAML, falling back to cloudpickle for
    non-representable leaves.
"""

import ast
import builtins
import copy
import importlib
import importlib.util
import os
import uuid
from collections import abc
from contextlib import contextmanager

__all__ = ["ConfigDict", "LazyCall", "LazyConfig", "locate"]


class ConfigDict(dict):
    """A dict with attribute access, the node type of every config tree.

    Stands in for omegaconf.DictConfig in the reference API surface
    (reference: libai/config/lazy.py:89-123 builds DictConfig nodes).
    """

    def __init__(self, *args, **kwargs):
        super().__init__()
        if args:
            (mapping,) = args
            for k, v in dict(mapping).items():
                self[k] = v
        for k, v in kwargs.items():
            self[k] = v

    @staticmethod
    def _convert(value):
        if isinstance(value, ConfigDict):
            return value
        if isinstance(value, dict):
            return ConfigDict(value)
        if isinstance(value, (list, tuple)):
            converted = [ConfigDict._convert(v) for v in value]
            return type(value)(converted) if isinstance(value, tuple) else converted
        return value

    def __setitem__(self, key, value):
        super().__setitem__(key, ConfigDict._convert(value))

    def __setattr__(self, key, value):
        self[key] = value

    def __getattr__(self, key):
        try:
            return self[key]
        except KeyError:
            raise AttributeError(
                f"ConfigDict has no key {key!r}; available: {list(self.keys())}"
            ) from None

    def __delattr__(self, key):
        try:
            del self[key]
        except KeyError:
            raise AttributeError(key) from None

    def update(self, *args, **kwargs):
        other = dict(*args, **kwargs)
        for k, v in other.items():
            self[k] = v

    def get(self, key, default=None):
        return super().get(key, default)

    def copy(self):
        return copy.deepcopy(self)

    def __deepcopy__(self, memo):
        out = ConfigDict()
        memo[id(self)] = out
        for k, v in self.items():
            dict.__setitem__(out, k, copy.deepcopy(v, memo))
        return out

    def __reduce__(self):
        # plain-dict payload keeps pickles stable across versions
        return (ConfigDict, (), None, None, iter(self.items()))

    def __setstate__(self, state):  # pragma: no cover - pickle protocol glue
        pass


def locate(name: str):
    """Import and return the object named by a dotted path.

    Mirrors the reference's ``locate`` used to resolve ``_target_`` strings
    (reference: libai/config/lazy.py dotted-path resolution).
    """
    if not name:
        raise ImportError("empty object name")
    parts = name.split(".")
    if hasattr(builtins, parts[0]) and len(parts) == 1:
        return getattr(builtins, parts[0])
    # longest importable module prefix, then getattr the rest
    for i in range(len(parts), 0, -1):
        module_name = ".".join(parts[:i])
        try:
            module = importlib.import_module(module_name)
        except ImportError:
            continue
        obj = module
        try:
            for attr in parts[i:]:
                obj = getattr(obj, attr)
        except AttributeError:
            raise ImportError(f"cannot locate {name!r}: no attribute {attr!r}") from None
        return obj
    raise ImportError(f"cannot locate {name!r}")


def _callable_to_path(target) -> str:
    if isinstance(target, str):
        return target
    module = getattr(target, "__module__", None)
    qualname = getattr(target, "__qualname__", None)
    if module is None or qualname is None or "<" in qualname:
        raise TypeError(f"cannot serialize target {target!r} to a dotted path")
    return f"{module}.{qualname}"


class LazyCall:
    """``LazyCall(cls)(**kw)`` -> ConfigDict with ``_target_`` set.

    Reference behavior: libai/config/lazy.py:89-123.
    """

    def __init__(self, target):
        if not (callable(target) or isinstance(target, str)):
            raise TypeError(f"LazyCall target must be callable or str, got {target!r}")
        self._target = target

    def __call__(self, **kwargs):
        node = ConfigDict(kwargs)
        # keep the live callable when possible (cheap + exact), store path for save()
        node["_target_"] = self._target
        return node


# ---------------------------------------------------------------------------
# Config-file loading with patched relative imports
# ---------------------------------------------------------------------------

_CFG_PACKAGE_NAME = "_libai_amd_cfg_loader"


def _random_package_name(filename):
    return _CFG_PACKAGE_NAME + str(uuid.uuid4())[:4] + "." + os.path.basename(filename)


@contextmanager
def _patch_import():
    """Allow ``from .common import train`` style imports between config files.

    Enables config composition across files located by path, the same idea as
    the reference's patched importer (reference: libai/config/lazy.py:168-224).
    """
    old_import = builtins.__import__

    def find_relative_file(original_file, relative_import_path, level):
        cur_file = os.path.dirname(original_file)
        for _ in range(level - 1):
            cur_file = os.path.dirname(cur_file)
        cur_name = relative_import_path.lstrip(".")
        for part in cur_name.split("."):
            cur_file = os.path.join(cur_file, part)
        if not cur_file.endswith(".py"):
            cur_file += ".py"
        if not os.path.isfile(cur_file):
            raise ImportError(
                f"cannot find config file {cur_file} for relative import "
                f"{'.' * level}{relative_import_path}"
            )
        return cur_file

    def new_import(name, globals=None, locals=None, fromlist=(), level=0):
        if (
            level != 0
            and globals is not None
            and (globals.get("__package__", "") or "").startswith(_CFG_PACKAGE_NAME)
        ):
            cur_file = find_relative_file(globals["__file__"], name, level)
            spec = importlib.machinery.ModuleSpec(
                _random_package_name(cur_file), None, origin=cur_file
            )
            module = importlib.util.module_from_spec(spec)
            module.__file__ = cur_file
            with open(cur_file) as f:
                content = f.read()
            exec(compile(content, cur_file, "exec"), module.__dict__)
            for attr in fromlist:
                if attr not in module.__dict__:
                    raise ImportError(
                        f"cannot import name {attr!r} from config file {cur_file}"
                    )
            return module
        return old_import(name, globals, locals, fromlist=fromlist, level=level)

    builtins.__import__ = new_import
    try:
        yield new_import
    finally:
        builtins.__import__ = old_import


class LazyConfig:
    """Load/save/override utilities for lazy configs.

    Reference behavior: libai/config/lazy.py:227-463.
    """

    @staticmethod
    def load(filename: str, keys=None):
        filename = filename.replace("/./", "/")
        if os.path.splitext(filename)[1] not in (".py", ".yaml", ".yml"):
            raise ValueError(f"config file {filename} has unsupported extension")
        if filename.endswith((".yaml", ".yml")):
            import yaml

            with open(filename) as f:
                obj = yaml.unsafe_load(f)
            ret = ConfigDict._convert(obj)
        else:
            with _patch_import():
                module_namespace = {
                    "__file__": os.path.abspath(filename),
                    "__package__": _random_package_name(filename),
                }
                with open(filename) as f:
                    content = f.read()
                exec(compile(content, filename, "exec"), module_namespace)
            ret = ConfigDict(
                {
                    name: value
                    for name, value in module_namespace.items()
                    if isinstance(value, (ConfigDict, dict, list))
                    and not name.startswith("_")
                }
            )
        if keys is None:
            return ret
        if isinstance(keys, str):
            return ret[keys]
        return tuple(ret[k] for k in keys)

    @staticmethod
    def load_rel(filename: str, keys=None):
        """Load a config file relative to the caller's file (reference parity)."""
        import inspect

        caller = inspect.stack()[1]
        caller_dir = os.path.dirname(os.path.abspath(caller.filename))
        return LazyConfig.load(os.path.join(caller_dir, filename), keys)

    # -- overrides ----------------------------------------------------------

    @staticmethod
    def apply_overrides(cfg, overrides):
        """Apply ``a.b.c=value`` overrides; values parse as Python literals.

        The reference uses hydra's override grammar (libai/config/lazy.py:361-401);
        we support the dotted assignment subset which is what LiBai configs use.
        """
        for override in overrides:
            if "=" not in override:
                raise ValueError(f"override {override!r} is not of the form key=value")
            key, value = override.split("=", 1)
            key = key.strip()
            try:
                parsed = ast.literal_eval(value)
            except (ValueError, SyntaxError):
                parsed = value  # bare string
            _set_dotted(cfg, key, parsed)
        return cfg

    # -- save / to_py -------------------------------------------------------

    @staticmethod
    def save(cfg, filename: str):
        """Save config to YAML; fall back to cloudpickle for exotic leaves.

        Reference: libai/config/lazy.py:303-359.
        """
        import yaml

        def sanitize(node):
            if isinstance(node, ConfigDict) or isinstance(node, dict):
                return {k: sanitize(v) for k, v in node.items()}
            if isinstance(node, (list, tuple)):
                return [sanitize(v) for v in node]
            if isinstance(node, (str, int, float, bool, type(None))):
                return node
            if callable(node):
                try:
                    return {"__callable__": _callable_to_path(node)}
                except TypeError:
                    return {"__repr__": repr(node)}
            return {"__repr__": repr(node)}

        try:
            with open(filename, "w") as f:
                yaml.safe_dump(sanitize(cfg), f, default_flow_style=False, sort_keys=False)
        except Exception:
            import cloudpickle

            with open(filename + ".pkl", "wb") as f:
                cloudpickle.dump(cfg, f)

    @staticmethod
    def to_py(cfg, prefix: str = "cfg.") -> str:
        """Pretty-print a config as executable-looking Python (reference parity)."""
        lines = []

        def fmt(node):
            if isinstance(node, dict) and "_target_" in node:
                target = node["_target_"]
                name = target if isinstance(target, str) else _callable_to_path(target)
                args = ", ".join(
                    f"{k}={fmt(v)}" for k, v in node.items() if k != "_target_"
                )
                return f"{name}({args})"
            if isinstance(node, dict):
                inner = ", ".join(f"{k!r}: {fmt(v)}" for k, v in node.items())
                return "{" + inner + "}"
            if isinstance(node, (list, tuple)):
                inner = ", ".join(fmt(v) for v in node)
                return ("[" + inner + "]") if isinstance(node, list) else "(" + inner + ")"
            return repr(node)

        if isinstance(cfg, dict):
            for k, v in cfg.items():
                lines.append(f"{prefix}{k} = {fmt(v)}")
        else:
            lines.append(f"{prefix.rstrip('.')} = {fmt(cfg)}")
        return "\n".join(lines)


def _set_dotted(cfg, dotted_key, value):
    parts = dotted_key.split(".")
    node = cfg
    for part in parts[:-1]:
        if isinstance(node, abc.Mapping):
            if part not in node:
                node[part] = ConfigDict()
            node = node[part]
        elif isinstance(node, (list, tuple)):
            node = node[int(part)]
        else:
            raise KeyError(f"cannot descend into {part!r} of override {dotted_key!r}")
    last = parts[-1]
    if isinstance(node, abc.Mapping):
        node[last] = value
    elif isinstance(node, list):
        node[int(last)] = value
    else:
        raise KeyError(f"cannot set {last!r} of override {dotted_key!r}")
