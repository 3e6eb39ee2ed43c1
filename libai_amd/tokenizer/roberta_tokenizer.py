"""RoBERTa tokenizer: GPT-2 byte BPE with BERT-style specials (reference:
libai/tokenizer/tokenization_roberta.py)."""

from .gpt2_tokenizer import GPT2Tokenizer

__all__ = ["RobertaTokenizer"]


class RobertaTokenizer(GPT2Tokenizer):
    def __init__(self, vocab_file, merges_file, bos_token="<s>", eos_token="</s>",
                 sep_token="</s>", cls_token="<s>", unk_token="<unk>",
                 pad_token="<pad>", mask_token="<mask>", **kwargs):
        super().__init__(vocab_file, merges_file, bos_token=bos_token,
                         eos_token=eos_token, unk_token=unk_token,
                         sep_token=sep_token, cls_token=cls_token,
                         pad_token=pad_token, mask_token=mask_token, **kwargs)

    def build_inputs_with_special_tokens(self, token_ids_0, token_ids_1=None):
        cls = [self._convert_token_to_id(self.cls_token)]
        sep = [self._convert_token_to_id(self.sep_token)]
        if token_ids_1 is None:
            return cls + token_ids_0 + sep
        return cls + token_ids_0 + sep + sep + token_ids_1 + sep
