"""ERNIE pretraining data: MLM masking + NSP sample building.

Reference: ppfleetx/data/dataset/ernie/dataset_utils.py (masked-LM span
selection, 80/10/10 replacement) and ernie_dataset.py:156-244
(build_training_sample: sentence pair + NSP label + masking). The
synthetic variant generates deterministic random "documents" so the full
pipeline runs without downloaded corpora.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import numpy as np
import torch
from torch.utils.data import Dataset


def create_masked_lm_predictions(
        tokens: np.ndarray, vocab_size: int, rng: np.random.RandomState,
        masked_lm_prob: float = 0.15, mask_token_id: int = 3,
        special_ids: Tuple[int, ...] = (0, 1, 2, 3),
        max_predictions: Optional[int] = None
        ) -> Tuple[np.ndarray, np.ndarray]:
    """80% [MASK] / 10% random / 10% keep; labels -1 where not predicted
    (dataset_utils.py create_masked_lm_predictions semantics)."""
    tokens = tokens.copy()
    labels = np.full_like(tokens, -1)
    cand = [i for i, t in enumerate(tokens) if t not in special_ids]
    rng.shuffle(cand)
    n_pred = max(1, int(round(len(cand) * masked_lm_prob)))
    if max_predictions is not None:
        n_pred = min(n_pred, max_predictions)
    for i in cand[:n_pred]:
        labels[i] = tokens[i]
        r = rng.rand()
        if r < 0.8:
            tokens[i] = mask_token_id
        elif r < 0.9:
            tokens[i] = rng.randint(len(special_ids), vocab_size)
        # else keep original
    return tokens, labels


class ErnieSyntheticDataset(Dataset):
    """Deterministic synthetic MLM+NSP samples:
    (input_ids, token_type_ids, masked_lm_labels, next_sentence_label)."""

    CLS, SEP, PAD, MASK = 1, 2, 0, 3

    def __init__(self, num_samples: int = 10000, seq_len: int = 512,
                 vocab_size: int = 40000, masked_lm_prob: float = 0.15,
                 mode: str = "Train", seed: int = 1234,
                 seq_cls: bool = False, num_classes: int = 2, **unused):
        self.num_samples = int(num_samples)
        self.seq_len = int(seq_len)
        self.vocab_size = int(vocab_size)
        self.masked_lm_prob = masked_lm_prob
        self.seed = seed
        # finetune form: (input_ids, token_type_ids, label) for
        # ErnieSeqClsModule (reference finetune_ernie yamls)
        self.seq_cls = bool(seq_cls)
        self.num_classes = int(num_classes)

    def __len__(self):
        return self.num_samples

    def __getitem__(self, idx):
        rng = np.random.RandomState(self.seed + idx)
        S = self.seq_len
        len_a = rng.randint(S // 4, S // 2)
        len_b = S - len_a - 3  # [CLS] a [SEP] b [SEP]
        sent_a = rng.randint(4, self.vocab_size, size=len_a)
        sent_b = rng.randint(4, self.vocab_size, size=len_b)
        nsp = int(rng.rand() < 0.5)  # 0 = continuation, 1 = random
        tokens = np.concatenate([[self.CLS], sent_a, [self.SEP], sent_b,
                                 [self.SEP]]).astype(np.int64)
        token_types = np.concatenate([
            np.zeros(len_a + 2, dtype=np.int64),
            np.ones(len_b + 1, dtype=np.int64)])
        if self.seq_cls:
            # separable synthetic task: the label is carried by the
            # parity of the first real token
            label = int(tokens[1] % self.num_classes)
            return (torch.from_numpy(tokens),
                    torch.from_numpy(token_types), torch.tensor(label))
        masked, labels = create_masked_lm_predictions(
            tokens, self.vocab_size, rng,
            masked_lm_prob=self.masked_lm_prob, mask_token_id=self.MASK,
            special_ids=(self.PAD, self.CLS, self.SEP, self.MASK))
        return (torch.from_numpy(masked), torch.from_numpy(token_types),
                torch.from_numpy(labels), torch.tensor(nsp))

def ernie_collate_fn(samples):
    n = len(samples[0])
    return tuple(torch.stack([s[i] for s in samples]) for i in range(n))


# build_dataloader picks this up as the dataset's collate
ErnieSyntheticDataset.collate_fn = staticmethod(ernie_collate_fn)


class ErnieWWMDataset(Dataset):
    """ERNIE pretraining over preprocessed Chinese whole-word-mask bins:
    `<prefix>_ids.npy` + `<prefix>_idx.npz` (+ `<prefix>_wwm.npy`
    continuation marks) written by tools/preprocess_data.py
    --model_family ernie --whole_word_mask. Masking selects WHOLE words
    (a token plus its '##' continuations) per
    ernie_preprocess.create_masked_lm_predictions_wwm — the reference's
    create_pretraining_data.py + dataset_utils.py pipeline."""

    CLS, SEP, PAD, MASK = 1, 2, 0, 3

    def __init__(self, input_prefix: str, seq_len: int = 512,
                 vocab_size: int = 40000, masked_lm_prob: float = 0.15,
                 mode: str = "Train", seed: int = 1234, **unused):
        self.ids = np.load(input_prefix + "_ids.npy", mmap_mode="r")
        lens = np.load(input_prefix + "_idx.npz")["lens"]
        self.doc_offsets = np.concatenate([[0], np.cumsum(lens)])
        import os as _os
        wwm_path = input_prefix + "_wwm.npy"
        self.wwm = np.load(wwm_path, mmap_mode="r") \
            if _os.path.exists(wwm_path) else None
        self.seq_len = int(seq_len)
        self.vocab_size = int(vocab_size)
        self.masked_lm_prob = masked_lm_prob
        self.seed = seed

    def __len__(self):
        return max(1, len(self.doc_offsets) - 1)

    def __getitem__(self, idx):
        idx = int(idx)
        rng = np.random.RandomState(self.seed + idx)
        lo, hi = int(self.doc_offsets[idx]), int(self.doc_offsets[idx + 1])
        body = np.asarray(self.ids[lo:hi], dtype=np.int64)
        cont = np.asarray(self.wwm[lo:hi], dtype=np.int64) \
            if self.wwm is not None else np.zeros_like(body)
        max_body = self.seq_len - 2  # [CLS] body [SEP]
        if len(body) > max_body:
            start = rng.randint(0, len(body) - max_body + 1)
            body = body[start:start + max_body]
            cont = cont[start:start + max_body]
            if cont[0]:  # don't start mid-word
                first = np.argmax(cont == 0)
                body, cont = body[first:], cont[first:]
        tokens = np.concatenate([[self.CLS], body, [self.SEP]])
        is_cont = np.concatenate([[0], cont, [0]])
        from paddlefleetx_amd.data.ernie_preprocess import \
            create_masked_lm_predictions_wwm
        masked, labels = create_masked_lm_predictions_wwm(
            tokens, is_cont, self.vocab_size, rng,
            masked_lm_prob=self.masked_lm_prob, mask_token_id=self.MASK,
            special_ids=(self.PAD, self.CLS, self.SEP, self.MASK))
        pad = self.seq_len - len(masked)
        masked = np.concatenate([masked, np.full(pad, self.PAD)])
        labels = np.concatenate([labels, np.full(pad, -1)])
        token_types = np.zeros(self.seq_len, dtype=np.int64)
        nsp = 0
        return (torch.from_numpy(masked.astype(np.int64)),
                torch.from_numpy(token_types),
                torch.from_numpy(labels.astype(np.int64)),
                torch.tensor(nsp))


ErnieWWMDataset.collate_fn = staticmethod(ernie_collate_fn)
