"""Dataset index-map builders.

Native C++ (pybind11) module `_index_map` built from csrc/index_builder.cpp
(re-implementation of the semantics of the reference's
data_tools/cpp/fast_index_map_helpers.cpp:32-193: weighted blending +
sequence packing over document boundaries); numpy fallback when the
extension is not built (slow path, same output).
"""

from __future__ import annotations

import numpy as np

try:
    from paddlefleetx_amd.data import _index_map as _cpp
except ImportError:
    _cpp = None


def build_sample_idx(doc_lens: np.ndarray, doc_idx: np.ndarray, seq_len: int,
                     num_epochs: int, tokens_per_epoch: int) -> np.ndarray:
    """Pack documents into seq_len+1-token samples.

    Returns int64 [num_samples+1, 2]: (doc_idx position, offset in doc).
    """
    if _cpp is not None:
        return _cpp.build_sample_idx(np.ascontiguousarray(doc_lens, dtype=np.int64),
                                     np.ascontiguousarray(doc_idx, dtype=np.int32),
                                     int(seq_len), int(num_epochs),
                                     int(tokens_per_epoch))
    return _build_sample_idx_np(doc_lens, doc_idx, seq_len, num_epochs,
                                tokens_per_epoch)


def _build_sample_idx_np(doc_lens, doc_idx, seq_len, num_epochs,
                         tokens_per_epoch) -> np.ndarray:
    """Canonical Megatron packing: sample i spans sample_idx[i]..sample_idx[i+1]
    inclusive of the boundary token (seq_len+1 tokens; the boundary token is
    shared between consecutive samples)."""
    num_samples = (num_epochs * tokens_per_epoch - 1) // seq_len
    sample_idx = np.zeros((num_samples + 1, 2), dtype=np.int64)
    si = 0
    di = 0   # index into doc_idx
    off = 0  # token offset in current doc
    sample_idx[si] = (di, off)
    si += 1
    while si <= num_samples:
        remaining = seq_len + 1
        while remaining != 0:
            if di >= len(doc_idx):
                return sample_idx[:si]
            doc_len = int(doc_lens[doc_idx[di]]) - off
            remaining -= doc_len
            if remaining <= 0:
                off += remaining + doc_len - 1
                remaining = 0
            else:
                di += 1
                off = 0
        sample_idx[si] = (di, off)
        si += 1
    return sample_idx


def build_blending_indices(weights: np.ndarray, num_samples: int):
    """Weighted multi-dataset blending (fast_index_map_helpers.cpp:32-90)."""
    if _cpp is not None:
        return _cpp.build_blending_indices(
            np.ascontiguousarray(weights, dtype=np.float64), int(num_samples))
    n = len(weights)
    dataset_index = np.zeros(num_samples, dtype=np.int8)
    dataset_sample_index = np.zeros(num_samples, dtype=np.int64)
    current = np.zeros(n, dtype=np.int64)
    for i in range(num_samples):
        errs = weights * (i + 1) - current
        d = int(np.argmax(errs))
        dataset_index[i] = d
        dataset_sample_index[i] = current[d]
        current[d] += 1
    return dataset_index, dataset_sample_index
