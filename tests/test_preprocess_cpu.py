"""Offline preprocessing pipeline: text -> jsonl -> token bins -> GPTDataset."""

import json
import os
import subprocess
import sys

import numpy as np
import pytest

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))


def _make_vocab_dir(tmp_path):
    from paddlefleetx_amd.data.tokenizers.gpt_tokenizer import bytes_to_unicode
    b2u = bytes_to_unicode()
    vocab = {c: i for i, c in enumerate(b2u.values())}
    vocab["<|endoftext|>"] = len(vocab)
    d = tmp_path / "vocab"
    d.mkdir()
    with open(d / "vocab.json", "w", encoding="utf-8") as f:
        json.dump(vocab, f, ensure_ascii=False)
    with open(d / "merges.txt", "w") as f:
        f.write("#version: 0.2\n")
    return str(d)


def test_full_preprocess_pipeline(tmp_path):
    vocab_dir = _make_vocab_dir(tmp_path)
    # raw text -> jsonl
    raw = tmp_path / "raw.txt"
    raw.write_text("\n".join(f"document number {i} with some text words"
                             for i in range(30)))
    jsonl = tmp_path / "corpus.jsonl"
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools/raw_trans_to_json.py"),
         "--input_path", str(raw), "--output_path", str(jsonl)],
        capture_output=True, text=True, cwd=REPO)
    assert r.returncode == 0, r.stderr
    assert len(jsonl.read_text().splitlines()) == 30

    # jsonl -> token bins
    prefix = tmp_path / "bins" / "corpus"
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools/preprocess_data.py"),
         "--input_path", str(jsonl), "--output_prefix", str(prefix),
         "--vocab_dir", vocab_dir, "--workers", "1"],
        capture_output=True, text=True, cwd=REPO)
    assert r.returncode == 0, r.stderr
    ids = np.load(str(prefix) + "_ids.npy")
    idx = np.load(str(prefix) + "_idx.npz")
    assert idx["lens"].sum() == len(ids)
    assert len(idx["lens"]) == 30

    # token bins -> GPTDataset samples
    from paddlefleetx_amd.data.gpt_dataset import GPTDataset
    ds = GPTDataset(str(tmp_path / "bins"), mode="Train", max_seq_len=32,
                    num_samples=20)
    tokens, pos, labels, mask = ds[0]
    assert tokens.shape == (32,)
    assert (labels[:-1] == tokens[1:]).all()  # packed shift-by-one


def test_build_pair_mapping_cpp_matches_python():
    """ERNIE/BERT sentence-pair mapping (reference
    fast_index_map_helpers.cpp:195-430): C++ and Python paths are
    bit-identical; samples respect min_num_sent and doc boundaries."""
    import numpy as np
    from paddlefleetx_amd.data import index_builder as ib
    docs = np.array([0, 3, 5, 10, 11], dtype=np.int64)
    sizes = np.array([30, 40, 50, 60, 20, 10, 10, 10, 10, 80, 5],
                     dtype=np.int32)
    args = (docs, sizes, 2, 100, 128, 0.3, 1234, 2)
    got = ib.build_pair_mapping(*args)
    saved = ib._cpp
    try:
        ib._cpp = None
        py = ib.build_pair_mapping(*args)
    finally:
        ib._cpp = saved
    assert np.array_equal(got, py)
    assert got.shape[1] == 3 and len(got) > 0
    # every sample stays inside one document and has >= 2 sentences
    bounds = list(zip(docs[:-1], docs[1:]))
    for s, e, t in got:
        assert e - s >= 2 and 4 <= t <= 128
        assert any(s >= lo and e <= hi for lo, hi in bounds)


def test_blended_dataset_ratio_and_provenance(tmp_path):
    """BlendedGPTDataset draws from every bin prefix in proportion to
    the weights (reference multi-dataset blending)."""
    d = tmp_path / "blend"
    d.mkdir()
    # two bins with distinguishable token values
    for name, val in (("aaa", 1), ("bbb", 2)):
        ids = np.full(2000, val, dtype=np.uint16)
        np.save(d / f"{name}_ids.npy", ids)
        np.savez(d / f"{name}_idx.npz",
                 lens=np.full(20, 100, dtype=np.int64))
    from paddlefleetx_amd.data.gpt_dataset import BlendedGPTDataset
    ds = BlendedGPTDataset(str(d), weights=[3.0, 1.0], mode="Train",
                           max_seq_len=16, num_samples=40)
    assert len(ds) == 40
    src = []
    for i in range(40):
        tokens, pos, labels, mask = ds[i]
        assert tokens.shape == (16,)
        vals = set(tokens.tolist())
        assert vals in ({1}, {2}), vals  # never mixes bins inside a sample
        src.append(tokens[0].item())
    # 3:1 weighting -> 30/10 exactly (deterministic greedy blending)
    assert src.count(1) == 30 and src.count(2) == 10


def test_blended_dataset_via_build_dataloader(tmp_path):
    """The blended dataset is addressable from the YAML Data section."""
    d = tmp_path / "blend2"
    d.mkdir()
    for name, val in (("x", 3), ("y", 4)):
        np.save(d / f"{name}_ids.npy", np.full(2000, val, dtype=np.uint16))
        np.savez(d / f"{name}_idx.npz",
                 lens=np.full(20, 100, dtype=np.int64))
    from paddlefleetx_amd.data import build_dataloader
    from paddlefleetx_amd.parallel.env import set_hcg
    from paddlefleetx_amd.parallel.topology import HybridTopology
    set_hcg(HybridTopology())
    cfg = {"Train": {"dataset": {"name": "BlendedGPTDataset",
                                 "input_dir": str(d),
                                 "weights": [1.0, 1.0],
                                 "max_seq_len": 16,
                                 "num_samples": 16},
                     "sampler": {"shuffle": False, "drop_last": True},
                     "loader": {"num_workers": 0}}}
    loader = build_dataloader(cfg, "Train", batch_size=4)
    tokens, pos, labels, mask = next(iter(loader))
    assert tokens.shape == (4, 16)
    assert set(tokens.unique().tolist()) <= {3, 4}
