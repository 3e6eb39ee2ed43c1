from .bert_tokenizer import BasicTokenizer, BertTokenizer, WordpieceTokenizer
from .build import build_tokenizer
from .gpt2_tokenizer import GPT2Tokenizer
from .roberta_tokenizer import RobertaTokenizer
from .t5_tokenizer import T5Tokenizer
from .tokenization_base import PreTrainedTokenizer

__all__ = [
    "PreTrainedTokenizer",
    "BertTokenizer",
    "BasicTokenizer",
    "WordpieceTokenizer",
    "GPT2Tokenizer",
    "RobertaTokenizer",
    "T5Tokenizer",
    "build_tokenizer",
]
