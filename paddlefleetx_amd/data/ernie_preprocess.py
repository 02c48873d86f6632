"""ERNIE Chinese offline preprocessing: word segmentation + whole-word
masking.

Reference: ppfleetx/data/data_tools/ernie/preprocess/
words_segmentation.py (jieba/LAC segmentation fns :64-101) and
create_pretraining_data.py:161 get_whole_word_mask_tokens. jieba/LAC are
not installed in this offline image, so the segmenter is pluggable:
`jieba` if importable, a greedy forward-maximum-matching segmenter over
a user lexicon, or per-character fallback — the downstream whole-word
mark algorithm is identical either way.
"""

from __future__ import annotations

import re
from typing import Callable, Iterable, List, Optional, Sequence

import numpy as np

__all__ = ["ChineseWordSegmenter", "get_whole_word_mask_tokens",
           "create_wwm_ids_and_marks", "create_masked_lm_predictions_wwm"]

_CJK_RE = re.compile(r"[一-龥]")


class ChineseWordSegmenter:
    """seg(text) -> list of words.

    backend "auto": jieba if importable, else lexicon FMM, else chars.
    backend "lexicon": greedy forward maximum matching over `lexicon`.
    backend "char": every CJK char is its own word (non-CJK runs kept).
    """

    def __init__(self, backend: str = "auto",
                 lexicon: Optional[Iterable[str]] = None,
                 max_word_len: int = 6):
        self.max_word_len = max_word_len
        self.lexicon = set(lexicon or [])
        self._jieba = None
        if backend == "auto":
            try:
                import jieba  # noqa: F401
                self._jieba = jieba
                backend = "jieba"
            except ImportError:
                backend = "lexicon" if self.lexicon else "char"
        elif backend == "jieba":
            import jieba
            self._jieba = jieba
        self.backend = backend

    def __call__(self, text: str) -> List[str]:
        if self.backend == "jieba":
            return [w for w in self._jieba.cut(text) if w.strip()]
        # split into CJK runs and non-CJK runs
        words: List[str] = []
        i = 0
        while i < len(text):
            ch = text[i]
            if _CJK_RE.match(ch):
                j = i
                while j < len(text) and _CJK_RE.match(text[j]):
                    j += 1
                words.extend(self._seg_cjk(text[i:j]))
                i = j
            else:
                j = i
                while j < len(text) and not _CJK_RE.match(text[j]):
                    j += 1
                w = text[i:j].strip()
                if w:
                    words.extend(w.split())
                i = j
        return words

    def _seg_cjk(self, run: str) -> List[str]:
        if self.backend != "lexicon" or not self.lexicon:
            return list(run)
        out: List[str] = []
        i = 0
        while i < len(run):
            for ln in range(min(self.max_word_len, len(run) - i), 1, -1):
                if run[i:i + ln] in self.lexicon:
                    out.append(run[i:i + ln])
                    i += ln
                    break
            else:
                out.append(run[i])
                i += 1
        return out


def get_whole_word_mask_tokens(tokens: Sequence[str], words: Sequence[str],
                               max_word_length: int = 6) -> List[str]:
    """Add '##' continuation marks to the non-initial characters of each
    segmented Chinese word so training-time masking can mask whole words
    (reference create_pretraining_data.py:161-226: same published
    algorithm — greedy longest word match against the segmentation)."""
    words_set = set(words)
    new_tokens: List[str] = []
    i = 0
    n = len(tokens)
    while i < n:
        if not _CJK_RE.search(tokens[i]):
            new_tokens.append(tokens[i])
            i += 1
            continue
        matched = False
        for length in range(max_word_length, 0, -1):
            if i + length > n:
                continue
            if "".join(tokens[i:i + length]) in words_set:
                new_tokens.append(tokens[i])
                new_tokens.extend("##" + tokens[i + k]
                                  for k in range(1, length))
                i += length
                matched = True
                break
        if not matched:
            new_tokens.append(tokens[i])
            i += 1
    return new_tokens


def create_wwm_ids_and_marks(text: str, tokenizer, segmenter=None):
    """text -> (ids, is_continuation) where is_continuation[i] = 1 marks
    a '##'-joined token (same information the reference persists through
    its vocab's ## entries)."""
    segmenter = segmenter or ChineseWordSegmenter()
    tokens = tokenizer.tokenize(text)
    plain = [t[2:] if t.startswith("##") else t for t in tokens]
    wwm = get_whole_word_mask_tokens(plain, segmenter(text))
    ids, cont = [], []
    for t in wwm:
        is_cont = t.startswith("##")
        tid = tokenizer.convert_tokens_to_ids(t)
        unk = getattr(tokenizer, "unk_token_id", None)
        if is_cont and unk is not None and tid == unk:
            # vocab may lack the ##-marked CJK form; keep the plain id
            tid = tokenizer.convert_tokens_to_ids(t[2:])
        ids.append(tid)
        cont.append(1 if is_cont else 0)
    return ids, cont


def create_masked_lm_predictions_wwm(
        tokens: np.ndarray, is_continuation: np.ndarray, vocab_size: int,
        rng: np.random.RandomState, masked_lm_prob: float = 0.15,
        mask_token_id: int = 3, special_ids=(0, 1, 2, 3),
        max_predictions: Optional[int] = None):
    """Whole-word MLM: candidate units are word spans (a token plus its
    '##' continuations); every token of a chosen span is masked together
    (reference dataset_utils.py whole-word branch semantics)."""
    tokens = tokens.copy()
    labels = np.full_like(tokens, -1)
    spans: List[List[int]] = []
    for i, t in enumerate(tokens):
        if int(t) in special_ids:
            continue
        if i > 0 and is_continuation[i] and spans:
            spans[-1].append(i)
        else:
            spans.append([i])
    order = rng.permutation(len(spans))
    budget = max(1, int(round(sum(len(s) for s in spans) * masked_lm_prob)))
    if max_predictions is not None:
        budget = min(budget, max_predictions)
    n_masked = 0
    for si in order:
        span = spans[si]
        if n_masked + len(span) > budget and n_masked > 0:
            continue
        r = rng.rand()  # one decision per WORD
        for i in span:
            labels[i] = tokens[i]
            if r < 0.8:
                tokens[i] = mask_token_id
            elif r < 0.9:
                tokens[i] = rng.randint(len(special_ids), vocab_size)
        n_masked += len(span)
        if n_masked >= budget:
            break
    return tokens, labels
