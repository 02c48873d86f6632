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


# ---------------------------------------------------------------------------
# TokenizerBase surface: batch encode / pad / truncate (reference
# tokenization_utils_base.py:622-1221 semantics)
# ---------------------------------------------------------------------------

def _toy_gpt():
    from paddlefleetx_amd.data.tokenizers.gpt_tokenizer import GPTTokenizer
    # byte-level vocab covering ascii letters+space via bytes_to_unicode
    from paddlefleetx_amd.data.tokenizers.gpt_tokenizer import \
        bytes_to_unicode
    b2u = bytes_to_unicode()
    vocab = {b2u[b]: i for i, b in enumerate(sorted(b2u))}
    vocab["<|endoftext|>"] = len(vocab)
    return GPTTokenizer(vocab, [])


def test_base_batch_encode_padding_longest():
    tok = _toy_gpt()
    enc = tok(["abc", "a"], padding=True, return_tensors="pt")
    assert enc["input_ids"].shape == enc["attention_mask"].shape
    assert enc["input_ids"].shape[1] == 3
    assert enc["attention_mask"][1].tolist() == [1, 0, 0]


def test_base_padding_max_length_and_multiple():
    tok = _toy_gpt()
    enc = tok.encode_plus("abc", padding="max_length", max_length=8)
    assert len(enc["input_ids"]) == 8
    assert enc["attention_mask"] == [1, 1, 1, 0, 0, 0, 0, 0]
    enc2 = tok.pad({"input_ids": [[1, 2, 3]], "attention_mask": [[1, 1, 1]]},
                   padding=True, pad_to_multiple_of=4)
    assert len(enc2["input_ids"][0]) == 4


def test_base_truncation_strategies():
    tok = _toy_gpt()
    long_ids = list(range(10))
    enc = tok.encode_plus(long_ids, truncation=True, max_length=6)
    assert enc["input_ids"] == long_ids[:6]
    # longest_first on a pair trims the longer side first
    enc = tok.encode_plus(list(range(8)), list(range(3)),
                          truncation="longest_first", max_length=7)
    assert len(enc["input_ids"]) == 7
    assert enc["input_ids"].count(2) == 2  # both sequences keep 0..2
    enc = tok.encode_plus(list(range(8)), list(range(3)),
                          truncation="only_first", max_length=7)
    assert enc["input_ids"] == [0, 1, 2, 3] + [0, 1, 2]


def test_base_token_type_ids_pair():
    import tempfile as _tf
    from paddlefleetx_amd.data.tokenizers.debertav2_tokenizer import \
        DebertaV2Tokenizer
    with _tf.TemporaryDirectory() as td:
        mp = _train_spm(td, ["[CLS]", "[SEP]", "[PAD]", "[MASK]"])
        tok = DebertaV2Tokenizer(mp)
        enc = tok.encode_plus("quick fox", "lazy dog",
                              return_token_type_ids=True)
        ids, tt = enc["input_ids"], enc["token_type_ids"]
        assert len(ids) == len(tt)
        sep1 = ids.index(tok.sep_token_id)
        assert all(t == 0 for t in tt[:sep1 + 1])
        assert all(t == 1 for t in tt[sep1 + 1:])
        # batch call with padding
        batch = tok(["quick fox", "the lazy dog jumps"], padding=True)
        lens = {len(r) for r in batch["input_ids"]}
        assert len(lens) == 1
