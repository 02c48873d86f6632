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


def _xs64(x: int) -> int:
    x ^= (x << 13) & 0xFFFFFFFFFFFFFFFF
    x ^= x >> 7
    x ^= (x << 17) & 0xFFFFFFFFFFFFFFFF
    return x & 0xFFFFFFFFFFFFFFFF


def build_pair_mapping(docs: np.ndarray, sizes: np.ndarray,
                       num_epochs: int, max_num_samples: int,
                       max_seq_length: int, short_seq_prob: float,
                       seed: int, min_num_sent: int = 2) -> np.ndarray:
    """BERT/ERNIE sentence-pair sample mapping
    (fast_index_map_helpers.cpp:195-430 semantics): greedy sentence
    packing per doc up to a per-sample target length; short targets with
    prob `short_seq_prob`. Bit-identical to the C++ path (same xorshift64
    stream)."""
    if _cpp is not None and hasattr(_cpp, "build_pair_mapping"):
        return _cpp.build_pair_mapping(
            np.ascontiguousarray(docs, dtype=np.int64),
            np.ascontiguousarray(sizes, dtype=np.int32), int(num_epochs),
            int(max_num_samples), int(max_seq_length), float(short_seq_prob),
            int(seed), int(min_num_sent))
    rng = seed if seed else 1
    short_cut = int(short_seq_prob * 18446744073709551615.0)
    out = []
    n_docs = len(docs) - 1

    def draw_target(r):
        target = max_seq_length
        if short_seq_prob > 0:
            r = _xs64(r)
            if r < short_cut:
                r2 = _xs64(r)
                target = 4 + (r2 % (max_seq_length - 3))
                return target, r2
        return target, r

    for _ in range(num_epochs):
        for doc in range(n_docs):
            if len(out) >= max_num_samples:
                break
            s0, s1 = int(docs[doc]), int(docs[doc + 1])
            if s1 - s0 < min_num_sent:
                continue
            start, tok, nsent = s0, 0, 0
            target, rng = draw_target(rng)
            for s in range(s0, s1):
                tok += int(sizes[s])
                nsent += 1
                last = s + 1 == s1
                if (tok >= target and nsent >= min_num_sent) or last:
                    if nsent >= min_num_sent:
                        out.append((start, s + 1, target))
                    start, tok, nsent = s + 1, 0, 0
                    target, rng = draw_target(rng)
                    if len(out) >= max_num_samples:
                        break
        if len(out) >= max_num_samples:
            break
    return np.asarray(out, dtype=np.int64).reshape(-1, 3)
