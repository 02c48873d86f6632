"""T5 sentencepiece tokenizer.

Reference: ppfleetx/data/tokenizers/t5_tokenizer.py (1,905 LoC wrapping a
sentencepiece model + HF-style specials). The sentencepiece package is
available offline; the model file must be local (no network).
"""

from __future__ import annotations

import os
from typing import List, Optional


from paddlefleetx_amd.data.tokenizers.tokenization_base import TokenizerBase


class T5Tokenizer(TokenizerBase):
    def __init__(self, sp_model_path: str, eos_token: str = "</s>",
                 unk_token: str = "<unk>", pad_token: str = "<pad>",
                 extra_ids: int = 100):
        import sentencepiece as spm
        self.sp = spm.SentencePieceProcessor()
        self.sp.Load(sp_model_path)
        TokenizerBase.__init__(self, eos_token=eos_token,
                               unk_token=unk_token, pad_token=pad_token)
        self.extra_ids = extra_ids

    # TokenizerBase hooks
    def _tokenize(self, text):
        return self.sp.EncodeAsPieces(text)

    def _convert_token_to_id(self, token):
        return self.sp.PieceToId(token)

    def _convert_id_to_token(self, idx):
        return self.sp.IdToPiece(int(idx))

    def convert_tokens_to_string(self, tokens):
        return self.sp.DecodePieces(tokens)

    def build_inputs_with_special_tokens(self, ids0, ids1=None):
        # T5 appends </s> to each sequence
        out = list(ids0) + [self.eos_token_id]
        if ids1 is not None:
            out += list(ids1) + [self.eos_token_id]
        return out

    @classmethod
    def from_pretrained(cls, path: str, **kw) -> "T5Tokenizer":
        if os.path.isdir(path):
            path = os.path.join(path, "spiece.model")
        return cls(path, **kw)

    @property
    def vocab_size(self) -> int:
        return self.sp.GetPieceSize() + self.extra_ids

    def encode(self, text: str, add_eos: bool = True, **kwargs) -> List[int]:
        if kwargs:
            return TokenizerBase.encode(self, text, **kwargs)
        ids = self.sp.EncodeAsIds(text)
        if add_eos:
            ids.append(self.eos_token_id)
        return ids

    def decode(self, ids, skip_special_tokens: bool = True) -> str:
        ids = [int(i) for i in ids if int(i) < self.sp.GetPieceSize()]
        return self.sp.DecodeIds(ids)

    def __len__(self):
        return self.vocab_size
