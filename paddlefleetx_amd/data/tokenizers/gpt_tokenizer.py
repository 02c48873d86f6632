"""GPT-2 byte-level BPE tokenizer.

Reference: ppfleetx/data/tokenizers/gpt_tokenizer.py (bytes_to_unicode :61,
GPTTokenizer :97, bpe :641, pad :481). Re-implemented from the algorithm;
vocab/merges are loaded from local files (this environment has no network,
so there is no downloader — pass explicit paths or use
`GPTTokenizer.from_pretrained(dir)` on a directory holding
`vocab.json` + `merges.txt`).
"""

from __future__ import annotations

import json
import os
import re
from functools import lru_cache
from typing import Dict, List, Optional, Tuple


@lru_cache()
def bytes_to_unicode() -> Dict[int, str]:
    """Reversible byte -> printable-unicode map (reference :61-84)."""
    bs = (list(range(ord("!"), ord("~") + 1)) +
          list(range(ord("\xa1"), ord("\xac") + 1)) +
          list(range(ord("\xae"), ord("\xff") + 1)))
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return dict(zip(bs, [chr(c) for c in cs]))


def get_pairs(word: Tuple[str, ...]):
    pairs = set()
    prev = word[0]
    for ch in word[1:]:
        pairs.add((prev, ch))
        prev = ch
    return pairs


from paddlefleetx_amd.data.tokenizers.tokenization_base import (
    BatchEncoding, TokenizerBase)


class GPTTokenizer(TokenizerBase):
    """Byte-level BPE with the GPT-2 regex pre-tokenizer; batch
    encode/pad/truncate surface from TokenizerBase."""

    # GPT-2 pre-tokenization pattern (reference :158)
    PAT = (r"'s|'t|'re|'ve|'m|'ll|'d| ?\p{L}+| ?\p{N}+|"
           r" ?[^\s\p{L}\p{N}]+|\s+(?!\S)|\s+")

    def __init__(self, vocab: Dict[str, int], merges: List[Tuple[str, str]],
                 errors: str = "replace", eos_token: str = "<|endoftext|>",
                 pad_token: Optional[str] = None):
        self.encoder = dict(vocab)
        self.decoder = {v: k for k, v in self.encoder.items()}
        self.byte_encoder = bytes_to_unicode()
        self.byte_decoder = {v: k for k, v in self.byte_encoder.items()}
        self.bpe_ranks = {tuple(m): i for i, m in enumerate(merges)}
        self.errors = errors
        self.cache: Dict[str, str] = {}
        try:
            import regex
            self._re = regex.compile(self.PAT)
        except ImportError:  # plain-re fallback (no \p classes)
            self._re = re.compile(r"\S+|\s+")
        TokenizerBase.__init__(self, eos_token=eos_token,
                               pad_token=pad_token or eos_token,
                               unk_token=eos_token)

    # -- constructors -------------------------------------------------------
    @classmethod
    def from_pretrained(cls, path: str, **kw) -> "GPTTokenizer":
        with open(os.path.join(path, "vocab.json"), encoding="utf-8") as f:
            vocab = json.load(f)
        merges: List[Tuple[str, str]] = []
        with open(os.path.join(path, "merges.txt"), encoding="utf-8") as f:
            for line in f:
                line = line.strip()
                if not line or line.startswith("#version"):
                    continue
                a, b = line.split()
                merges.append((a, b))
        return cls(vocab, merges, **kw)

    @classmethod
    def gpt2_tokenizer(cls) -> "GPTTokenizer":
        """Load gpt2 vocab from well-known local paths (no network)."""
        for cand in (os.environ.get("GPT2_VOCAB_DIR", ""),
                     os.path.expanduser("~/.cache/gpt2"),
                     "/root/data/gpt2"):
            if cand and os.path.exists(os.path.join(cand, "vocab.json")):
                return cls.from_pretrained(cand)
        raise FileNotFoundError(
            "gpt2 vocab.json/merges.txt not found; set GPT2_VOCAB_DIR "
            "(no network available to download them)")

    # -- BPE ---------------------------------------------------------------
    def bpe(self, token: str) -> str:
        if token in self.cache:
            return self.cache[token]
        word = tuple(token)
        pairs = get_pairs(word) if len(word) > 1 else None
        if not pairs:
            return token
        while True:
            bigram = min(pairs, key=lambda p: self.bpe_ranks.get(p, 1 << 30))
            if bigram not in self.bpe_ranks:
                break
            first, second = bigram
            new_word: List[str] = []
            i = 0
            while i < len(word):
                try:
                    j = word.index(first, i)
                except ValueError:
                    new_word.extend(word[i:])
                    break
                new_word.extend(word[i:j])
                i = j
                if i < len(word) - 1 and word[i] == first and \
                        word[i + 1] == second:
                    new_word.append(first + second)
                    i += 2
                else:
                    new_word.append(word[i])
                    i += 1
            word = tuple(new_word)
            if len(word) == 1:
                break
            pairs = get_pairs(word)
        out = " ".join(word)
        self.cache[token] = out
        return out

    # -- public API ---------------------------------------------------------
    def tokenize(self, text: str) -> List[str]:
        bpe_tokens: List[str] = []
        for token in self._re.findall(text):
            token = "".join(self.byte_encoder[b]
                            for b in token.encode("utf-8"))
            bpe_tokens.extend(self.bpe(token).split(" "))
        return bpe_tokens

    # TokenizerBase hooks
    def _tokenize(self, text: str) -> List[str]:
        return self.tokenize(text)

    def _convert_token_to_id(self, token: str) -> int:
        return self.encoder[token]

    def _convert_id_to_token(self, idx: int) -> str:
        return self.decoder[int(idx)]

    def convert_tokens_to_string(self, tokens: List[str]) -> str:
        data = bytearray(self.byte_decoder[c] for c in "".join(tokens))
        return data.decode("utf-8", errors=self.errors)

    def encode(self, text: str, **kwargs) -> List[int]:
        if not kwargs:
            return self.convert_tokens_to_ids(self.tokenize(text))
        return TokenizerBase.encode(self, text, **kwargs)

    def decode(self, ids, skip_special_tokens: bool = False) -> str:
        return TokenizerBase.decode(self, ids, skip_special_tokens)

    def __len__(self) -> int:
        return len(self.encoder)

    @property
    def vocab_size(self) -> int:
        return len(self.encoder)

    def pad(self, batch_ids, max_length: Optional[int] = None,
            pad_to_multiple_of: Optional[int] = None, **kw):
        """Right-padding (reference :481-520). Accepts the legacy
        list-of-id-lists OR a BatchEncoding (TokenizerBase surface)."""
        if isinstance(batch_ids, dict):
            return TokenizerBase.pad(self, batch_ids,
                                     max_length=max_length,
                                     pad_to_multiple_of=pad_to_multiple_of,
                                     **kw)
        longest = max(len(x) for x in batch_ids)
        target = max_length or longest
        if pad_to_multiple_of:
            target = ((target + pad_to_multiple_of - 1) //
                      pad_to_multiple_of) * pad_to_multiple_of
        out, mask = [], []
        for ids in batch_ids:
            padn = target - len(ids)
            out.append(list(ids) + [self.pad_token_id] * padn)
            mask.append([1] * len(ids) + [0] * padn)
        return {"input_ids": out, "attention_mask": mask}
