"""T5 SentencePiece tokenizer (reference: libai/tokenizer/tokenization_t5.py)."""

import os

from .tokenization_base import PreTrainedTokenizer

__all__ = ["T5Tokenizer"]


class T5Tokenizer(PreTrainedTokenizer):
    vocab_files_names = {"vocab_file": "spiece.model"}

    def __init__(self, vocab_file, eos_token="</s>", unk_token="<unk>",
                 pad_token="<pad>", extra_ids=100, **kwargs):
        extra = [f"<extra_id_{i}>" for i in range(extra_ids)]
        addl = kwargs.pop("additional_special_tokens", None) or extra
        super().__init__(eos_token=eos_token, unk_token=unk_token,
                         pad_token=pad_token, additional_special_tokens=addl,
                         **kwargs)
        import sentencepiece as spm

        self.vocab_file = vocab_file
        self.sp_model = spm.SentencePieceProcessor()
        self.sp_model.Load(vocab_file)
        self._extra_ids = extra_ids

    @property
    def vocab_size(self):
        return self.sp_model.get_piece_size() + self._extra_ids

    def get_vocab(self):
        v = {self._convert_id_to_token(i): i for i in range(self.vocab_size)}
        return v

    def _tokenize(self, text):
        return self.sp_model.EncodeAsPieces(text)

    def _convert_token_to_id(self, token):
        if token.startswith("<extra_id_"):
            num = int(token[len("<extra_id_"):-1])
            return self.vocab_size - num - 1
        return self.sp_model.piece_to_id(token)

    def _convert_id_to_token(self, index):
        if index < self.sp_model.get_piece_size():
            return self.sp_model.IdToPiece(index)
        return f"<extra_id_{self.vocab_size - 1 - index}>"

    def convert_tokens_to_string(self, tokens):
        return self.sp_model.decode_pieces(
            [t for t in tokens if not t.startswith("<extra_id_")]
        )

    def build_inputs_with_special_tokens(self, token_ids_0, token_ids_1=None):
        eos = [self._convert_token_to_id(self.eos_token)]
        if token_ids_1 is None:
            return token_ids_0 + eos
        return token_ids_0 + eos + token_ids_1 + eos

    def save_vocabulary(self, save_directory):
        import shutil

        out = os.path.join(save_directory, "spiece.model")
        if os.path.abspath(self.vocab_file) != os.path.abspath(out):
            shutil.copyfile(self.vocab_file, out)
        return (out,)
