from .bert_model import BertForPreTraining, BertModel
from .bloom import BloomForCausalLM, BloomModel
from .clip import CLIPModel
from .contrastive import MoCoV3, SimCSEModel
from .build import build_model
from .gpt_model import GPTForPreTraining, GPTModel
from .llama import LlamaForCausalLM, LlamaModel
from .convnext import ConvNeXt
from .mae import MAEForPreTraining
from .palm import PaLMForCausalLM, PaLMModel
from .resmlp import ResMLP
from .roberta_model import RobertaForCausalLM, RobertaForPreTraining, RobertaModel
from .swin_transformer import SwinTransformer
from .swin_transformer_v2 import SwinTransformerV2
from .t5_model import T5ForPreTraining, T5Model
from .vision_transformer import VisionTransformer

__all__ = [
    "GPTModel",
    "GPTForPreTraining",
    "BertModel",
    "BertForPreTraining",
    "T5Model",
    "T5ForPreTraining",
    "RobertaModel",
    "RobertaForPreTraining",
    "RobertaForCausalLM",
    "LlamaModel",
    "LlamaForCausalLM",
    "VisionTransformer",
    "SwinTransformer",
    "SwinTransformerV2",
    "ResMLP",
    "BloomModel",
    "BloomForCausalLM",
    "ConvNeXt",
    "MAEForPreTraining",
    "SimCSEModel",
    "MoCoV3",
    "CLIPModel",
    "PaLMModel",
    "PaLMForCausalLM",
    "build_model",
]
