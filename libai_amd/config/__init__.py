from .arguments import default_argument_parser
from .config import configurable, get_config, try_get_key
from .instantiate import dump_dataclass, instantiate
from .lazy import ConfigDict, LazyCall, LazyConfig, locate

__all__ = [
    "ConfigDict",
    "LazyCall",
    "LazyConfig",
    "locate",
    "instantiate",
    "dump_dataclass",
    "configurable",
    "try_get_key",
    "get_config",
    "default_argument_parser",
]
