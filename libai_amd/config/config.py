"""``configurable`` decorator and config helpers.

Reference behavior: libai/config/config.py:32-198.
"""

import functools
import inspect

from .lazy import ConfigDict

__all__ = ["configurable", "try_get_key", "get_config"]


def configurable(init_func=None, *, from_config=None):
    """Decorate a class ``__init__`` or a function so it can be called either
    with explicit arguments or with a config object routed through
    ``from_config`` (reference: libai/config/config.py:32-168).

    Usage on a class::

        class Foo:
            @configurable
            def __init__(self, a, b=2): ...
            @classmethod
            def from_config(cls, cfg): return {"a": cfg.A, "b": cfg.B}

    Usage on a function::

        @configurable(from_config=lambda cfg: {"a": cfg.A})
        def build(a, b=2): ...
    """
    if init_func is not None:
        assert (
            inspect.isfunction(init_func)
            and from_config is None
            and init_func.__name__ == "__init__"
        ), "Incorrect use of @configurable: decorate __init__ or use from_config="

        @functools.wraps(init_func)
        def wrapped(self, *args, **kwargs):
            try:
                from_config_func = type(self).from_config
            except AttributeError as e:
                raise AttributeError(
                    "Class with @configurable __init__ must define a from_config classmethod"
                ) from e
            if not inspect.ismethod(from_config_func):
                raise TypeError("from_config must be a classmethod")
            if _called_with_cfg(*args, **kwargs):
                explicit = _get_args_from_config(from_config_func, *args, **kwargs)
                init_func(self, **explicit)
            else:
                init_func(self, *args, **kwargs)

        return wrapped

    if from_config is None:
        raise TypeError("configurable() must be given from_config when used on a function")
    assert inspect.isfunction(from_config), "from_config must be a function"

    def wrapper(orig_func):
        @functools.wraps(orig_func)
        def wrapped(*args, **kwargs):
            if _called_with_cfg(*args, **kwargs):
                explicit = _get_args_from_config(from_config, *args, **kwargs)
                return orig_func(**explicit)
            return orig_func(*args, **kwargs)

        wrapped.from_config = from_config
        return wrapped

    return wrapper


def _called_with_cfg(*args, **kwargs):
    if args and isinstance(args[0], (dict, ConfigDict)):
        return True
    if isinstance(kwargs.pop("cfg", None), (dict, ConfigDict)):
        return True
    return False


def _get_args_from_config(from_config_func, *args, **kwargs):
    signature = inspect.signature(from_config_func)
    params = list(signature.parameters.keys())
    # drop the implicit cls for classmethods
    first = params[0] if params else None
    if first == "cls":
        params = params[1:]
    supports_var_kw = any(
        p.kind == inspect.Parameter.VAR_KEYWORD for p in signature.parameters.values()
    )
    if supports_var_kw:
        ret = from_config_func(*args, **kwargs)
    else:
        supported, extra = {}, {}
        for name, value in kwargs.items():
            (supported if name in params else extra)[name] = value
        ret = from_config_func(*args, **supported)
        ret.update(extra)
    return ret


def try_get_key(cfg, *names, default=None):
    """Return the first present dotted key among ``names`` (reference parity:
    libai/config/config.py:171)."""
    for name in names:
        node = cfg
        found = True
        for part in name.split("."):
            if isinstance(node, dict) and part in node:
                node = node[part]
            else:
                found = False
                break
        if found:
            return node
    return default


def get_config(config_path: str):
    """Load one of the in-repo default configs by path relative to ``configs/``."""
    import os

    from .lazy import LazyConfig

    root = os.path.join(os.path.dirname(__file__), "..", "..", "configs")
    path = os.path.normpath(os.path.join(root, config_path))
    if not os.path.isfile(path):
        raise FileNotFoundError(f"no builtin config {config_path!r} (looked at {path})")
    return LazyConfig.load(path)
