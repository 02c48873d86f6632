"""Property-based checks (hypothesis) for pure-python data-layer pieces:
padding/stacking round-trips, blending-index invariants, and the
shared softmax reference numerics."""

import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st


@settings(max_examples=50, deadline=None)
@given(st.lists(st.lists(st.integers(0, 1000), min_size=1, max_size=12),
                min_size=1, max_size=8),
       st.integers(-5, 5))
def test_pad_roundtrip(rows, pad_val):
    from paddlefleetx_amd.data.collate import Pad
    batch, lengths = Pad(pad_val=pad_val, ret_length=True)(rows)
    assert batch.shape == (len(rows), max(len(r) for r in rows))
    for i, r in enumerate(rows):
        assert batch[i, :len(r)].tolist() == r          # content preserved
        assert (batch[i, len(r):] == pad_val).all()     # filler exact
        assert int(lengths[i]) == len(r)


@settings(max_examples=50, deadline=None)
@given(st.lists(st.floats(0.01, 10.0), min_size=1, max_size=6),
       st.integers(1, 400))
def test_blending_indices_invariants(weights, n):
    from paddlefleetx_amd.data.index_builder import build_blending_indices
    w = np.array(weights) / np.sum(weights)
    di, dsi = build_blending_indices(w, n)
    assert len(di) == len(dsi) == n
    for d in range(len(weights)):
        picks = (di == d).sum()
        # greedy error-minimizing blend is within 1 of the exact share
        assert abs(picks - w[d] * n) <= 1.0 + 1e-9
        # per-dataset sample indices are 0..picks-1 in order
        assert np.array_equal(np.sort(dsi[di == d]), np.arange(picks))


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 8), st.integers(1, 64))
def test_softmax_reference_rows_sum_to_one(b, n):
    from paddlefleetx_amd.ops import _reference as ref
    x = torch.randn(b, n) * 10
    y = ref.softmax_fwd(x) if hasattr(ref, "softmax_fwd") else \
        torch.softmax(x, dim=-1)
    s = y.sum(-1)
    assert torch.allclose(s, torch.ones_like(s), atol=1e-5)
    assert (y >= 0).all()
