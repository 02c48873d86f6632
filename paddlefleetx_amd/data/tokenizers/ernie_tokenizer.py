"""ERNIE (BERT-style WordPiece) tokenizer.

Reference: ppfleetx/data/tokenizers/ernie_tokenizer.py (which wraps
paddlenlp's ErnieTokenizer). Standard public algorithm: basic tokenizer
(whitespace + punctuation split, CJK chars isolated, optional
lowercase) followed by greedy longest-match WordPiece with '##'
continuation marks. Vocab is a local `vocab.txt` (one token per line).
"""

from __future__ import annotations

import os
import unicodedata
from typing import Dict, List, Optional

from paddlefleetx_amd.data.tokenizers.tokenization_base import TokenizerBase


def _is_cjk(cp: int) -> bool:
    return (0x4E00 <= cp <= 0x9FFF or 0x3400 <= cp <= 0x4DBF or
            0x20000 <= cp <= 0x2A6DF or 0xF900 <= cp <= 0xFAFF)


def _is_punct(ch: str) -> bool:
    cp = ord(ch)
    if (33 <= cp <= 47) or (58 <= cp <= 64) or (91 <= cp <= 96) or \
            (123 <= cp <= 126):
        return True
    return unicodedata.category(ch).startswith("P")


class BasicTokenizer:
    def __init__(self, do_lower_case: bool = True):
        self.do_lower_case = do_lower_case

    def tokenize(self, text: str) -> List[str]:
        if self.do_lower_case:
            text = text.lower()
        out: List[str] = []
        buf: List[str] = []

        def flush():
            if buf:
                out.append("".join(buf))
                buf.clear()

        for ch in text:
            if ch.isspace():
                flush()
            elif _is_cjk(ord(ch)) or _is_punct(ch):
                flush()
                out.append(ch)
            else:
                buf.append(ch)
        flush()
        return out


class WordpieceTokenizer:
    def __init__(self, vocab: Dict[str, int], unk_token: str = "[UNK]",
                 max_input_chars_per_word: int = 100):
        self.vocab = vocab
        self.unk_token = unk_token
        self.max_chars = max_input_chars_per_word

    def tokenize(self, word: str) -> List[str]:
        if len(word) > self.max_chars:
            return [self.unk_token]
        out: List[str] = []
        start = 0
        while start < len(word):
            end = len(word)
            cur = None
            while start < end:
                sub = word[start:end]
                if start > 0:
                    sub = "##" + sub
                if sub in self.vocab:
                    cur = sub
                    break
                end -= 1
            if cur is None:
                return [self.unk_token]
            out.append(cur)
            start = end
        return out


class ErnieTokenizer(TokenizerBase):
    def __init__(self, vocab: Dict[str, int], do_lower_case: bool = True,
                 cls_token: str = "[CLS]", sep_token: str = "[SEP]",
                 pad_token: str = "[PAD]", mask_token: str = "[MASK]",
                 unk_token: str = "[UNK]"):
        TokenizerBase.__init__(self, cls_token=cls_token,
                               sep_token=sep_token, pad_token=pad_token,
                               mask_token=mask_token, unk_token=unk_token)
        self.vocab = dict(vocab)
        self.inv_vocab = {v: k for k, v in self.vocab.items()}
        self.basic = BasicTokenizer(do_lower_case)
        self.wordpiece = WordpieceTokenizer(self.vocab, unk_token)

    @classmethod
    def from_pretrained(cls, path: str, **kw) -> "ErnieTokenizer":
        if os.path.isdir(path):
            path = os.path.join(path, "vocab.txt")
        vocab: Dict[str, int] = {}
        with open(path, encoding="utf-8") as f:
            for i, line in enumerate(f):
                vocab[line.rstrip("\n")] = i
        return cls(vocab, **kw)

    @property
    def vocab_size(self) -> int:
        return len(self.vocab)

    def __len__(self):
        return len(self.vocab)

    def _tokenize(self, text: str) -> List[str]:
        out: List[str] = []
        for word in self.basic.tokenize(text):
            out.extend(self.wordpiece.tokenize(word))
        return out

    def _convert_token_to_id(self, token: str) -> int:
        return self.vocab.get(token, self.vocab.get(self.unk_token, 0))

    def _convert_id_to_token(self, idx: int) -> str:
        return self.inv_vocab.get(int(idx), self.unk_token)

    def convert_tokens_to_string(self, tokens: List[str]) -> str:
        return " ".join(tokens).replace(" ##", "")

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
