"""Sentencepiece-backed tokenizers (T5, DeBERTa-v2) — trained on a tiny
local corpus (offline; no downloaded models in this environment)."""

import os
import tempfile

import pytest


def _train_spm(tmpdir, specials):
    spm = pytest.importorskip("sentencepiece")
    corpus = os.path.join(tmpdir, "c.txt")
    with open(corpus, "w") as f:
        for line in ("the quick brown fox jumps over the lazy dog",
                     "pack my box with five dozen liquor jugs",
                     "how vexingly quick daft zebras jump") * 20:
            f.write(line + "\n")
    prefix = os.path.join(tmpdir, "m")
    spm.SentencePieceTrainer.Train(
        input=corpus, model_prefix=prefix, vocab_size=45,
        user_defined_symbols=list(specials))
    return prefix + ".model"


def test_t5_tokenizer_roundtrip():
    from paddlefleetx_amd.data.tokenizers.t5_tokenizer import T5Tokenizer
    with tempfile.TemporaryDirectory() as td:
        mp = _train_spm(td, ["</s>", "<pad>"])
        tok = T5Tokenizer(mp)
        ids = tok.encode("the quick brown fox")
        assert ids[-1] == tok.eos_token_id
        assert "quick" in tok.decode(ids)


def test_debertav2_tokenizer_pair_encoding():
    from paddlefleetx_amd.data.tokenizers.debertav2_tokenizer import \
        DebertaV2Tokenizer
    with tempfile.TemporaryDirectory() as td:
        mp = _train_spm(td, ["[CLS]", "[SEP]", "[PAD]", "[MASK]"])
        tok = DebertaV2Tokenizer(mp)
        ids = tok.encode("the quick fox", "lazy dog")
        assert ids[0] == tok.cls_token_id
        assert ids.count(tok.sep_token_id) == 2
        assert "quick" in tok.decode(ids)
        single = tok.encode("five dozen jugs", add_special_tokens=False)
        assert tok.cls_token_id not in single
