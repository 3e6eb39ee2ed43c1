from .gpt_model import GPTForPreTraining, GPTModel
from .build import build_model

__all__ = ["GPTModel", "GPTForPreTraining", "build_model"]
