"""GPT pretraining datasets.

- GPTDataset: mmap token bins (`*_ids.npy` + `*_idx.npz`) with doc/sample/
  shuffle index construction (reference dataset/gpt_dataset.py:42-480;
  index build in the native module paddlefleetx_amd.data.index_builder,
  mirroring data_tools/cpp/fast_index_map_helpers.cpp:92 build_sample_idx).
- GPTSyntheticDataset: random tokens of the same sample structure
  (tokens, position_ids, labels, loss_mask) for benchmarks — no network,
  no corpora (BASELINE.json: synthetic data, random-init weights).
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import numpy as np
import torch
from torch.utils.data import Dataset

from paddlefleetx_amd.utils.log import logger


class GPTSyntheticDataset(Dataset):
    """Random-token dataset shaped like GPTDataset samples."""

    def __init__(self, num_samples: int = 10000, seq_len: int = 1024,
                 vocab_size: int = 50304, seed: int = 1234, **unused):
        self.num_samples = num_samples
        self.seq_len = seq_len
        self.vocab_size = vocab_size
        self.seed = seed

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        rng = np.random.RandomState((self.seed + idx) % (2 ** 31))
        ids = rng.randint(0, self.vocab_size, size=self.seq_len + 1, dtype=np.int64)
        tokens = torch.from_numpy(ids[:-1].copy())
        labels = torch.from_numpy(ids[1:].copy())
        position_ids = torch.arange(self.seq_len, dtype=torch.int64)
        loss_mask = torch.ones(self.seq_len, dtype=torch.float32)
        return tokens, position_ids, labels, loss_mask


def _build_doc_idx(documents: np.ndarray, num_epochs: int,
                   rng: np.random.RandomState) -> np.ndarray:
    """Shuffled document order repeated num_epochs times (gpt_dataset.py:413)."""
    doc_idx = np.tile(documents, num_epochs)
    rng.shuffle(doc_idx)
    return doc_idx.astype(np.int32)


def _build_shuffle_idx(num_samples: int, total: int,
                       rng: np.random.RandomState) -> np.ndarray:
    """Two-part shuffle (gpt_dataset.py:466)."""
    dtype = np.int64 if total >= 2 ** 31 else np.int32
    first = np.arange(num_samples, dtype=dtype)
    rng.shuffle(first)
    last = np.arange(num_samples, total, dtype=dtype)
    rng.shuffle(last)
    return np.concatenate((first, last))


class GPTDataset(Dataset):
    """Token-bin dataset with sequence packing across document boundaries."""

    def __init__(self, input_dir: str, split=(949, 50, 1), mode: str = "Train",
                 max_seq_len: int = 1024, num_samples: int = 10000,
                 seed: int = 1234, prefix: str = None, **unused):
        if prefix is None:
            files = sorted(f for f in os.listdir(input_dir)
                           if f.endswith("_ids.npy"))
            assert files, f"no *_ids.npy token bins under {input_dir}"
            prefix = os.path.join(input_dir, files[0][:-len("_ids.npy")])
        self.token_ids = np.load(prefix + "_ids.npy", mmap_mode="r")
        idx = np.load(prefix + "_idx.npz")
        # lens[i] = tokens in doc i; docs[i] = start offset of doc i
        lens = idx["lens"] if "lens" in idx else idx["arr_0"]
        self.doc_offsets = np.concatenate(([0], np.cumsum(lens))).astype(np.int64)
        num_docs = len(lens)

        # 949/50/1 style split over documents (gpt_dataset.py split logic)
        splits = np.array(split, dtype=np.float64)
        splits = splits / splits.sum()
        bounds = (np.cumsum(splits) * num_docs).astype(np.int64)
        lo = {"Train": 0, "Eval": bounds[0], "Test": bounds[1]}[mode]
        hi = {"Train": bounds[0], "Eval": bounds[1], "Test": bounds[2]}[mode]
        documents = np.arange(lo, hi, dtype=np.int32)

        self.seq_len = max_seq_len
        rng = np.random.RandomState(seed)
        tokens_per_epoch = int(lens[documents].sum())
        num_epochs = max(1, int(np.ceil(
            (num_samples * (max_seq_len + 1)) / max(1, tokens_per_epoch))))
        self.doc_idx = _build_doc_idx(documents, num_epochs, rng)
        from paddlefleetx_amd.data.index_builder import build_sample_idx
        self.sample_idx = build_sample_idx(
            lens.astype(np.int64), self.doc_idx.astype(np.int32),
            max_seq_len, num_epochs, tokens_per_epoch)
        n_avail = self.sample_idx.shape[0] - 1
        self.shuffle_idx = _build_shuffle_idx(min(num_samples, n_avail),
                                              n_avail, rng)
        self.num_samples = min(num_samples, n_avail)
        logger.info(f"GPTDataset[{mode}]: {num_docs} docs, "
                    f"{self.num_samples} samples of seq {max_seq_len}")

    def __len__(self):
        return self.num_samples

    def _get_tokens(self, idx: int) -> np.ndarray:
        idx = int(self.shuffle_idx[idx])
        doc_f, off_f = self.sample_idx[idx]
        doc_l, off_l = self.sample_idx[idx + 1]
        if doc_f == doc_l:
            start = self.doc_offsets[self.doc_idx[doc_f]] + off_f
            return np.array(self.token_ids[start:start + (off_l - off_f) + 1],
                            dtype=np.int64)
        parts = []
        d0 = self.doc_offsets[self.doc_idx[doc_f]]
        parts.append(self.token_ids[d0 + off_f:self.doc_offsets[self.doc_idx[doc_f] + 1]])
        for d in range(doc_f + 1, doc_l):
            s = self.doc_offsets[self.doc_idx[d]]
            parts.append(self.token_ids[s:self.doc_offsets[self.doc_idx[d] + 1]])
        dl = self.doc_offsets[self.doc_idx[doc_l]]
        parts.append(self.token_ids[dl:dl + off_l + 1])
        return np.concatenate(parts).astype(np.int64)

    def __getitem__(self, idx):
        toks = self._get_tokens(idx)
        assert len(toks) == self.seq_len + 1, (len(toks), self.seq_len)
        tokens = torch.from_numpy(toks[:-1].copy())
        labels = torch.from_numpy(toks[1:].copy())
        position_ids = torch.arange(self.seq_len, dtype=torch.int64)
        loss_mask = torch.ones(self.seq_len, dtype=torch.float32)
        return tokens, position_ids, labels, loss_mask


class BlendedGPTDataset(Dataset):
    """Weighted blend over every token-bin prefix under `input_dir`
    (reference multi-dataset blending, fast_index_map_helpers.cpp:32-90
    via data_tools blending): sample i comes from dataset
    `dataset_index[i]`, drawn in proportion to `weights`."""

    def __init__(self, input_dir: str, weights=None, split=(949, 50, 1),
                 mode: str = "Train", max_seq_len: int = 1024,
                 num_samples: int = 10000, seed: int = 1234, **unused):
        from paddlefleetx_amd.data.index_builder import build_blending_indices
        files = sorted(f for f in os.listdir(input_dir)
                       if f.endswith("_ids.npy"))
        assert files, f"no *_ids.npy token bins under {input_dir}"
        prefixes = [os.path.join(input_dir, f[:-len("_ids.npy")])
                    for f in files]
        if weights is None:
            weights = [1.0] * len(prefixes)
        assert len(weights) == len(prefixes), \
            f"{len(weights)} weights for {len(prefixes)} bin prefixes"
        w = np.asarray(weights, dtype=np.float64)
        w = w / w.sum()
        self.children = []
        for p, wi in zip(prefixes, w):
            # oversample each child slightly (reference 1.005 margin) so
            # the blend never runs a child dry
            n_i = int(np.ceil(num_samples * float(wi) * 1.005)) + 1
            self.children.append(GPTDataset(
                input_dir, split=split, mode=mode, max_seq_len=max_seq_len,
                num_samples=n_i, seed=seed, prefix=p))
        self.dataset_index, self.dataset_sample_index = \
            build_blending_indices(w, int(num_samples))
        self.num_samples = int(num_samples)
        self.seq_len = max_seq_len
        logger.info(f"BlendedGPTDataset[{mode}]: {len(prefixes)} bins, "
                    f"weights {np.round(w, 3).tolist()}, "
                    f"{num_samples} samples")

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        child = self.children[int(self.dataset_index[idx])]
        return child[int(self.dataset_sample_index[idx]) % len(child)]
