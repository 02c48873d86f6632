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


from paddlefleetx_amd.data.tokenizers.tokenization_base import TokenizerBase


class DebertaV2Tokenizer(TokenizerBase):
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
        TokenizerBase.__init__(self, cls_token=cls_token,
                               sep_token=sep_token, unk_token=unk_token,
                               pad_token=pad_token, mask_token=mask_token)

    @classmethod
    def from_pretrained(cls, path: str, **kw) -> "DebertaV2Tokenizer":
        if os.path.isdir(path):
            path = os.path.join(path, "spm.model")
        return cls(path, **kw)

    @property
    def vocab_size(self) -> int:
        return max(self.sp.GetPieceSize(),
                   max(self._special.values()) + 1)

    # TokenizerBase hooks
    def _tokenize(self, text: str) -> List[str]:
        if self.do_lower_case:
            text = text.lower()
        return self.sp.EncodeAsPieces(text)

    def _convert_token_to_id(self, token):
        if token in self._special:
            return self._special[token]
        return self.sp.PieceToId(token)

    def _convert_id_to_token(self, idx):
        idx = int(idx)
        for tok, i in self._special.items():
            if i == idx:
                return tok
        return self.sp.IdToPiece(idx)

    def convert_tokens_to_string(self, tokens):
        return self.sp.DecodePieces(
            [t for t in tokens if t not in self._special])

    def build_inputs_with_special_tokens(self, ids0, ids1=None):
        out = [self.cls_token_id] + list(ids0) + [self.sep_token_id]
        if ids1 is not None:
            out += list(ids1) + [self.sep_token_id]
        return out

    def create_token_type_ids_from_sequences(self, ids0, ids1=None):
        out = [0] * (len(ids0) + 2)
        if ids1 is not None:
            out += [1] * (len(ids1) + 1)
        return out

    def tokenize(self, text: str) -> List[str]:
        return self._tokenize(text)

    def encode(self, text: str, text_pair: Optional[str] = None,
               add_special_tokens: bool = True, **kwargs) -> List[int]:
        if kwargs:
            return TokenizerBase.encode(self, text, text_pair=text_pair,
                                        add_special_tokens=add_special_tokens,
                                        **kwargs)
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

    def decode(self, ids, skip_special_tokens: bool = True) -> str:
        keep = [int(i) for i in ids
                if int(i) < self.sp.GetPieceSize()
                and int(i) not in self._special.values()]
        return self.sp.DecodeIds(keep)

    def __len__(self):
        return self.vocab_size
