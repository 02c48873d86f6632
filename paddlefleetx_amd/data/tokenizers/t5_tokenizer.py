"""T5 sentencepiece tokenizer.

Reference: ppfleetx/data/tokenizers/t5_tokenizer.py (1,905 LoC wrapping a
sentencepiece model + HF-style specials). The sentencepiece package is
available offline; the model file must be local (no network).
"""

from __future__ import annotations

import os
from typing import List, Optional


class T5Tokenizer:
    def __init__(self, sp_model_path: str, eos_token: str = "</s>",
                 unk_token: str = "<unk>", pad_token: str = "<pad>",
                 extra_ids: int = 100):
        import sentencepiece as spm
        self.sp = spm.SentencePieceProcessor()
        self.sp.Load(sp_model_path)
        self.eos_token, self.unk_token, self.pad_token = (eos_token,
                                                          unk_token,
                                                          pad_token)
        self.extra_ids = extra_ids
        self.eos_token_id = self.sp.PieceToId(eos_token)
        self.unk_token_id = self.sp.PieceToId(unk_token)
        self.pad_token_id = self.sp.PieceToId(pad_token)

    @classmethod
    def from_pretrained(cls, path: str, **kw) -> "T5Tokenizer":
        if os.path.isdir(path):
            path = os.path.join(path, "spiece.model")
        return cls(path, **kw)

    @property
    def vocab_size(self) -> int:
        return self.sp.GetPieceSize() + self.extra_ids

    def encode(self, text: str, add_eos: bool = True) -> List[int]:
        ids = self.sp.EncodeAsIds(text)
        if add_eos:
            ids.append(self.eos_token_id)
        return ids

    def decode(self, ids: List[int]) -> str:
        ids = [int(i) for i in ids if int(i) < self.sp.GetPieceSize()]
        return self.sp.DecodeIds(ids)

    def __len__(self):
        return self.vocab_size
