from .weight_init import init_method_normal, scaled_init_method_normal

__all__ = ["init_method_normal", "scaled_init_method_normal"]
from .model_loader import (
    BertLoaderHuggerFace,
    GPT2LoaderHuggerFace,
    LlamaLoaderHuggerFace,
    ModelLoaderHuggerFace,
    ModelLoaderLiBai,
    RobertaLoaderHuggerFace,
    SwinLoaderHuggerFace,
    SwinV2LoaderHuggerFace,
    ViTLoaderHuggerFace,
)

__all__ += [
    "ModelLoaderLiBai",
    "ModelLoaderHuggerFace",
    "GPT2LoaderHuggerFace",
    "BertLoaderHuggerFace",
    "RobertaLoaderHuggerFace",
    "LlamaLoaderHuggerFace",
    "ViTLoaderHuggerFace",
    "SwinLoaderHuggerFace",
    "SwinV2LoaderHuggerFace",
]
