"""Offline eval datasets: WikiText-style LM PPL + LAMBADA cloze.

Reference: ppfleetx/data/dataset/gpt_dataset.py — LM_Eval_Dataset :484
(sliding window with overlapping_eval; loss only on the non-overlap
tail; info = [num_original_tokens, num_tokenized_tokens]) and
Lambada_Eval_Dataset :589 (prefix + target word; cloze accuracy over the
target positions).

Token sources: a raw `tokens` array (tests/synthetic) or `eval_path`
pointing at a .npy int array (offline-tokenized text).
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import numpy as np
import torch
from torch.utils.data import Dataset


def _load_tokens(eval_path: Optional[str], tokens) -> np.ndarray:
    if tokens is not None:
        return np.asarray(tokens, dtype=np.int64)
    if eval_path is None:
        raise ValueError("need eval_path or tokens")
    if eval_path.endswith(".npy"):
        return np.load(eval_path).astype(np.int64)
    raise ValueError(f"unsupported eval file {eval_path} (expect .npy tokens;"
                     " tokenize text offline with tools/preprocess)")


class LM_Eval_Dataset(Dataset):
    """Overlapping-window LM evaluation (gpt_dataset.py:484-586)."""

    def __init__(self, eval_path: Optional[str] = None, tokens=None,
                 max_seq_len: int = 1024, overlapping_eval: int = 32,
                 num_original_tokens: Optional[int] = None,
                 pad_idx: int = 0, mode: str = "Eval", **unused):
        self.tokens = _load_tokens(eval_path, tokens)
        self.seq_len = max_seq_len
        self.pad_idx = pad_idx
        self.overlapping_eval = overlapping_eval or max_seq_len
        self.num_tokenized_tokens = len(self.tokens)
        self.num_original_tokens = num_original_tokens or len(self.tokens)
        targets = max(len(self.tokens) - self.seq_len, 0)
        self.total_seq = max(targets // self.overlapping_eval + 1, 1)

    def __len__(self):
        return self.total_seq

    def __getitem__(self, idx):
        start = idx * self.overlapping_eval
        end = start + self.seq_len
        chunk = self.tokens[start:end + 1]
        toks = chunk[:-1]
        labels = chunk[1:]
        real = len(toks)
        npad = self.seq_len - real
        loss_mask = np.ones(real, dtype=np.float32)
        if idx != 0 and self.overlapping_eval < self.seq_len:
            # only score the fresh tail (gpt_dataset.py:575-580)
            loss_mask[:max(0, real - self.overlapping_eval)] = 0.0
        if npad > 0:
            toks = np.concatenate([toks, np.full(npad, self.pad_idx)])
            labels = np.concatenate([labels, np.full(npad, self.pad_idx)])
            loss_mask = np.concatenate([loss_mask, np.zeros(npad,
                                                            dtype=np.float32)])
        position_ids = np.arange(self.seq_len, dtype=np.int64)
        info = np.array([self.num_original_tokens, self.num_tokenized_tokens],
                        dtype=np.int64)
        return (torch.from_numpy(toks.astype(np.int64)),
                torch.from_numpy(position_ids),
                torch.from_numpy(labels.astype(np.int64)),
                torch.from_numpy(loss_mask), torch.from_numpy(info))


class Lambada_Eval_Dataset(Dataset):
    """Cloze: predict the final word's tokens (gpt_dataset.py:589-650).

    samples: list of (prefix_tokens, target_tokens) or eval_path to an
    .npy object array of token lists where the LAST word's tokens are the
    target.
    """

    def __init__(self, eval_path: Optional[str] = None,
                 samples: Optional[Sequence] = None, max_seq_len: int = 1024,
                 pad_idx: int = 0, mode: str = "Eval", **unused):
        if samples is None:
            arr = np.load(eval_path, allow_pickle=True)
            samples = [(list(x[0]), list(x[1])) for x in arr]
        self.samples = list(samples)
        self.seq_len = max_seq_len
        self.pad_idx = pad_idx

    def __len__(self):
        return len(self.samples)

    def __getitem__(self, idx):
        prefix, target = self.samples[idx]
        toks = list(prefix) + list(target)
        toks = toks[-(self.seq_len + 1):]
        labels = toks[1:]
        toks = toks[:-1]
        loss_mask = np.zeros(len(toks), dtype=np.float32)
        loss_mask[-len(target):] = 1.0
        npad = self.seq_len - len(toks)
        if npad > 0:
            toks = toks + [self.pad_idx] * npad
            labels = labels + [self.pad_idx] * npad
            loss_mask = np.concatenate([loss_mask,
                                        np.zeros(npad, dtype=np.float32)])
        position_ids = np.arange(self.seq_len, dtype=np.int64)
        info = np.array([len(self.samples), 0], dtype=np.int64)
        return (torch.tensor(toks, dtype=torch.long),
                torch.from_numpy(position_ids),
                torch.tensor(labels, dtype=torch.long),
                torch.from_numpy(loss_mask), torch.from_numpy(info))


def eval_collate_fn(samples):
    return tuple(torch.stack([s[i] for s in samples]) for i in range(5))


LM_Eval_Dataset.collate_fn = staticmethod(eval_collate_fn)
Lambada_Eval_Dataset.collate_fn = staticmethod(eval_collate_fn)
