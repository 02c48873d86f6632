"""DeBERTa-v2 sentencepiece tokenizer.

Reference: ppfleetx/data/tokenizers/deberta_v2_tokenizer.py (2,163 LoC
wrapping a sentencepiece model with BERT-style specials and pair
encoding). Offline counterpart: sentencepiece is importable; the model
file must be local. Used for the Imagen DebertaV2 text-conditioning path
(run_text2im_64x64_DebertaV2 config family).
"""

from __future__ import annotations

import os
from typing import List, Optional


class DebertaV2Tokenizer:
    def __init__(self, sp_model_path: str, cls_token: str = "[CLS]",
                 sep_token: str = "[SEP]", unk_token: str = "[UNK]",
                 pad_token: str = "[PAD]", mask_token: str = "[MASK]",
                 do_lower_case: bool = False):
        import sentencepiece as spm
        self.sp = spm.SentencePieceProcessor()
        self.sp.Load(sp_model_path)
        self.do_lower_case = do_lower_case
        n = self.sp.GetPieceSize()
        # BERT-style specials occupy the conventional low ids when the
        # spm model doesn't define them
        self._special = {}
        for i, tok in enumerate((pad_token, cls_token, sep_token,
                                 unk_token, mask_token)):
            pid = self.sp.PieceToId(tok)
            self._special[tok] = pid if pid != self.sp.unk_id() or \
                tok == unk_token else n + i
        self.cls_token, self.sep_token = cls_token, sep_token
        self.unk_token, self.pad_token = unk_token, pad_token
        self.mask_token = mask_token

    @classmethod
    def from_pretrained(cls, path: str, **kw) -> "DebertaV2Tokenizer":
        if os.path.isdir(path):
            path = os.path.join(path, "spm.model")
        return cls(path, **kw)

    @property
    def cls_token_id(self):
        return self._special[self.cls_token]

    @property
    def sep_token_id(self):
        return self._special[self.sep_token]

    @property
    def pad_token_id(self):
        return self._special[self.pad_token]

    @property
    def mask_token_id(self):
        return self._special[self.mask_token]

    @property
    def vocab_size(self) -> int:
        return max(self.sp.GetPieceSize(),
                   max(self._special.values()) + 1)

    def tokenize(self, text: str) -> List[str]:
        if self.do_lower_case:
            text = text.lower()
        return self.sp.EncodeAsPieces(text)

    def encode(self, text: str, text_pair: Optional[str] = None,
               add_special_tokens: bool = True) -> List[int]:
        ids = self.sp.EncodeAsIds(
            text.lower() if self.do_lower_case else text)
        if not add_special_tokens:
            return ids
        out = [self.cls_token_id] + ids + [self.sep_token_id]
        if text_pair is not None:
            pids = self.sp.EncodeAsIds(
                text_pair.lower() if self.do_lower_case else text_pair)
            out += pids + [self.sep_token_id]
        return out

    def decode(self, ids: List[int]) -> str:
        keep = [int(i) for i in ids
                if int(i) < self.sp.GetPieceSize()
                and int(i) not in self._special.values()]
        return self.sp.DecodeIds(keep)

    def __len__(self):
        return self.vocab_size
