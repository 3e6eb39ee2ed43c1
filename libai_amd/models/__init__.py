from .bert_model import BertForPreTraining, BertModel
from .build import build_model
from .gpt_model import GPTForPreTraining, GPTModel
from .llama import LlamaForCausalLM, LlamaModel
from .vision_transformer import VisionTransformer

__all__ = [
    "GPTModel",
    "GPTForPreTraining",
    "BertModel",
    "BertForPreTraining",
    "LlamaModel",
    "LlamaForCausalLM",
    "VisionTransformer",
    "build_model",
]
