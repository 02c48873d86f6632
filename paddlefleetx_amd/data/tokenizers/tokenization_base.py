"""Tokenizer base utilities: batch encode / pad / truncate surface.

Compact re-implementation of the reference's HF-style base
(ppfleetx/data/tokenizers/tokenization_utils_base.py:622 BatchEncoding,
:1199 TruncationStrategy, :1211 PaddingStrategy, :1222
SpecialTokensMixin, plus the encode/encode_plus/batch_encode_plus/pad
methods of the PreTrainedTokenizerBase further down the same file).
The GPT/T5/DeBERTa tokenizers subclass `TokenizerBase` and only provide
`_tokenize`, `_convert_token_to_id`, `_convert_id_to_token` and
`convert_tokens_to_string`.
"""

from __future__ import annotations

from enum import Enum
from typing import Any, Dict, List, Optional, Sequence, Union

__all__ = ["BatchEncoding", "PaddingStrategy", "TruncationStrategy",
           "SpecialTokensMixin", "TokenizerBase"]


class PaddingStrategy(str, Enum):
    LONGEST = "longest"
    MAX_LENGTH = "max_length"
    DO_NOT_PAD = "do_not_pad"


class TruncationStrategy(str, Enum):
    ONLY_FIRST = "only_first"
    ONLY_SECOND = "only_second"
    LONGEST_FIRST = "longest_first"
    DO_NOT_TRUNCATE = "do_not_truncate"


class BatchEncoding(dict):
    """dict of encoder outputs with attribute access and
    `convert_to_tensors` (reference BatchEncoding :622-1197)."""

    def __getattr__(self, item):
        try:
            return self[item]
        except KeyError:
            raise AttributeError(item)

    def convert_to_tensors(self, tensor_type: str = "pt") -> "BatchEncoding":
        if tensor_type in ("pt", "torch"):
            import torch
            for k, v in self.items():
                if not isinstance(v, torch.Tensor):
                    self[k] = torch.tensor(v)
        elif tensor_type == "np":
            import numpy as np
            for k, v in self.items():
                self[k] = np.asarray(v)
        else:
            raise ValueError(f"unknown tensor type {tensor_type}")
        return self


class SpecialTokensMixin:
    """Named special tokens + registration (reference :1222-1533)."""

    SPECIAL_TOKENS_ATTRIBUTES = ["bos_token", "eos_token", "unk_token",
                                 "sep_token", "pad_token", "cls_token",
                                 "mask_token"]

    def __init__(self, **kwargs):
        for name in self.SPECIAL_TOKENS_ATTRIBUTES:
            setattr(self, name, kwargs.pop(name, None))
        self.additional_special_tokens: List[str] = list(
            kwargs.pop("additional_special_tokens", []) or [])

    @property
    def all_special_tokens(self) -> List[str]:
        toks = [getattr(self, n) for n in self.SPECIAL_TOKENS_ATTRIBUTES]
        return [t for t in toks if t is not None] \
            + list(self.additional_special_tokens)

    @property
    def all_special_ids(self) -> List[int]:
        return [self.convert_tokens_to_ids(t)
                for t in self.all_special_tokens]

    def _special_id(self, name: str) -> Optional[int]:
        tok = getattr(self, name, None)
        if tok is None:
            return None
        return self.convert_tokens_to_ids(tok)

    @property
    def bos_token_id(self):
        return self._special_id("bos_token")

    @property
    def eos_token_id(self):
        return self._special_id("eos_token")

    @property
    def unk_token_id(self):
        return self._special_id("unk_token")

    @property
    def sep_token_id(self):
        return self._special_id("sep_token")

    @property
    def pad_token_id(self):
        return self._special_id("pad_token")

    @property
    def cls_token_id(self):
        return self._special_id("cls_token")

    @property
    def mask_token_id(self):
        return self._special_id("mask_token")


class TokenizerBase(SpecialTokensMixin):
    """encode / encode_plus / batch_encode_plus / __call__ / pad with the
    reference's padding + truncation semantics."""

    model_max_length: int = 10 ** 9

    # --- hooks for subclasses -------------------------------------------
    def _tokenize(self, text: str) -> List[str]:
        raise NotImplementedError

    def _convert_token_to_id(self, token: str) -> int:
        raise NotImplementedError

    def _convert_id_to_token(self, idx: int) -> str:
        raise NotImplementedError

    def convert_tokens_to_string(self, tokens: List[str]) -> str:
        return " ".join(tokens)

    # --- conversion ------------------------------------------------------
    def tokenize(self, text: str) -> List[str]:
        return self._tokenize(text)

    def convert_tokens_to_ids(self, tokens):
        if tokens is None:
            return None
        if isinstance(tokens, str):
            return self._convert_token_to_id(tokens)
        return [self._convert_token_to_id(t) for t in tokens]

    def convert_ids_to_tokens(self, ids):
        if isinstance(ids, int):
            return self._convert_id_to_token(ids)
        return [self._convert_id_to_token(i) for i in ids]

    # --- special-token composition (override for CLS/SEP models) --------
    def build_inputs_with_special_tokens(self, ids0: List[int],
                                         ids1: Optional[List[int]] = None
                                         ) -> List[int]:
        if ids1 is None:
            return list(ids0)
        return list(ids0) + list(ids1)

    def create_token_type_ids_from_sequences(
            self, ids0: List[int], ids1: Optional[List[int]] = None
    ) -> List[int]:
        if ids1 is None:
            return [0] * len(self.build_inputs_with_special_tokens(ids0))
        return [0] * len(ids0) + [1] * len(ids1)

    def num_special_tokens_to_add(self, pair: bool = False) -> int:
        a, b = [0], [1]
        return len(self.build_inputs_with_special_tokens(
            a, b if pair else None)) - len(a) - (len(b) if pair else 0)

    # --- truncation ------------------------------------------------------
    def truncate_sequences(self, ids0: List[int],
                           ids1: Optional[List[int]],
                           num_tokens_to_remove: int,
                           strategy: TruncationStrategy):
        """Reference truncate_sequences semantics: longest_first removes
        one token at a time from the currently-longer sequence."""
        if num_tokens_to_remove <= 0:
            return ids0, ids1
        if strategy == TruncationStrategy.ONLY_FIRST or ids1 is None:
            return ids0[:max(0, len(ids0) - num_tokens_to_remove)], ids1
        if strategy == TruncationStrategy.ONLY_SECOND:
            return ids0, ids1[:max(0, len(ids1) - num_tokens_to_remove)]
        ids0, ids1 = list(ids0), list(ids1)
        for _ in range(num_tokens_to_remove):
            if len(ids0) >= len(ids1) and ids0:
                ids0.pop()
            elif ids1:
                ids1.pop()
        return ids0, ids1

    # --- single-pair encode ---------------------------------------------
    def encode_plus(self, text: Union[str, List[int]],
                    text_pair: Optional[Union[str, List[int]]] = None,
                    add_special_tokens: bool = True,
                    padding: Union[bool, str] = False,
                    truncation: Union[bool, str] = False,
                    max_length: Optional[int] = None,
                    return_attention_mask: bool = True,
                    return_token_type_ids: bool = False,
                    return_tensors: Optional[str] = None) -> BatchEncoding:
        ids0 = self._to_ids(text)
        ids1 = self._to_ids(text_pair) if text_pair is not None else None

        trunc = self._trunc_strategy(truncation)
        if trunc != TruncationStrategy.DO_NOT_TRUNCATE:
            limit = max_length or self.model_max_length
            n_special = self.num_special_tokens_to_add(ids1 is not None) \
                if add_special_tokens else 0
            total = len(ids0) + (len(ids1) if ids1 else 0) + n_special
            ids0, ids1 = self.truncate_sequences(ids0, ids1, total - limit,
                                                 trunc)
        if add_special_tokens:
            seq = self.build_inputs_with_special_tokens(ids0, ids1)
            type_ids = self.create_token_type_ids_from_sequences(ids0, ids1)
        else:
            seq = list(ids0) + (list(ids1) if ids1 else [])
            type_ids = [0] * len(ids0) + [1] * (len(ids1) if ids1 else 0)

        enc = BatchEncoding(input_ids=seq)
        if return_token_type_ids:
            enc["token_type_ids"] = type_ids
        if return_attention_mask:
            enc["attention_mask"] = [1] * len(seq)
        enc = self.pad(enc, padding=padding, max_length=max_length)
        if return_tensors:
            enc.convert_to_tensors(return_tensors)
        return enc

    def encode(self, text, **kwargs) -> List[int]:
        kwargs.setdefault("return_attention_mask", False)
        return self.encode_plus(text, **kwargs)["input_ids"]

    def batch_encode_plus(self, batch: Sequence, **kwargs) -> BatchEncoding:
        padding = kwargs.pop("padding", False)
        return_tensors = kwargs.pop("return_tensors", None)
        max_length = kwargs.pop("max_length", None)
        encs = []
        for item in batch:
            if isinstance(item, tuple):
                e = self.encode_plus(item[0], item[1], padding=False,
                                     max_length=max_length, **kwargs)
            else:
                e = self.encode_plus(item, padding=False,
                                     max_length=max_length, **kwargs)
            encs.append(e)
        merged = BatchEncoding(
            {k: [e[k] for e in encs] for k in encs[0].keys()})
        merged = self.pad(merged, padding=padding, max_length=max_length)
        if return_tensors:
            merged.convert_to_tensors(return_tensors)
        return merged

    def __call__(self, text, text_pair=None, **kwargs):
        if isinstance(text, (list, tuple)) and text and \
                isinstance(text[0], (str, tuple, list)) and not \
                isinstance(text[0], int):
            if text_pair is not None:
                batch = list(zip(text, text_pair))
            else:
                batch = list(text)
            return self.batch_encode_plus(batch, **kwargs)
        return self.encode_plus(text, text_pair, **kwargs)

    # --- padding ---------------------------------------------------------
    def pad(self, encoding: BatchEncoding,
            padding: Union[bool, str] = True,
            max_length: Optional[int] = None,
            pad_to_multiple_of: Optional[int] = None) -> BatchEncoding:
        strat = self._pad_strategy(padding)
        if strat == PaddingStrategy.DO_NOT_PAD:
            return encoding
        pad_id = self.pad_token_id
        if pad_id is None:
            pad_id = 0
        batched = encoding["input_ids"] and \
            isinstance(encoding["input_ids"][0], list)
        rows = encoding["input_ids"] if batched else [encoding["input_ids"]]
        if strat == PaddingStrategy.MAX_LENGTH:
            target = max_length or self.model_max_length
        else:
            target = max(len(r) for r in rows)
        if pad_to_multiple_of:
            target = ((target + pad_to_multiple_of - 1)
                      // pad_to_multiple_of * pad_to_multiple_of)

        def pad_rows(key, value):
            vrows = value if batched else [value]
            out = []
            for r in vrows:
                fill = pad_id if key == "input_ids" else 0
                out.append(list(r) + [fill] * (target - len(r)))
            return out if batched else out[0]

        for k in list(encoding.keys()):
            encoding[k] = pad_rows(k, encoding[k])
        return encoding

    # --- decode ----------------------------------------------------------
    def decode(self, ids, skip_special_tokens: bool = False) -> str:
        if hasattr(ids, "tolist"):
            ids = ids.tolist()
        toks = self.convert_ids_to_tokens(ids)
        if skip_special_tokens:
            sp = set(self.all_special_tokens)
            toks = [t for t in toks if t not in sp]
        return self.convert_tokens_to_string(toks)

    # --- helpers ---------------------------------------------------------
    def _to_ids(self, text) -> List[int]:
        if isinstance(text, str):
            return self.convert_tokens_to_ids(self.tokenize(text))
        return list(text)

    @staticmethod
    def _pad_strategy(padding) -> PaddingStrategy:
        if padding is True or padding == "longest":
            return PaddingStrategy.LONGEST
        if padding == "max_length":
            return PaddingStrategy.MAX_LENGTH
        return PaddingStrategy.DO_NOT_PAD

    @staticmethod
    def _trunc_strategy(truncation) -> TruncationStrategy:
        if truncation is True or truncation == "longest_first":
            return TruncationStrategy.LONGEST_FIRST
        if truncation in ("only_first", "only_second"):
            return TruncationStrategy(truncation)
        return TruncationStrategy.DO_NOT_TRUNCATE
