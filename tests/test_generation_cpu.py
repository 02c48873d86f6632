"""Generation path: KV-cache decode, logits processors, top-k/top-p, tokenizer."""

import pytest
import torch

from paddlefleetx_amd.models.gpt.generation import (GPTForGeneration,
                                                    TopKProcess, TopPProcess)
from paddlefleetx_amd.models.gpt.model import GPTModel
from paddlefleetx_amd.models.gpt.processor import (
    MinLengthLogitsProcessor, RepetitionPenaltyLogitsProcessor,
    get_logits_processor)
from paddlefleetx_amd.parallel.env import init_dist_env


@pytest.fixture(scope="module", autouse=True)
def _env():
    init_dist_env({"Distributed": {}})


def tiny_gpt(vocab=128, hidden=64, layers=2, heads=4, maxpos=64):
    torch.manual_seed(0)
    return GPTModel(vocab_size=vocab, hidden_size=hidden, num_layers=layers,
                    num_attention_heads=heads, max_position_embeddings=maxpos,
                    hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
                    fused_attn=False)


def test_kv_cache_matches_full_forward():
    """Incremental decode with cache == full forward logits."""
    gpt = tiny_gpt().eval()
    gen = GPTForGeneration(gpt, {"max_dec_len": 4})
    ids = torch.randint(0, 128, (2, 10))
    pos = torch.arange(10).unsqueeze(0).expand(2, -1)
    with torch.no_grad():
        # full forward over 10 tokens
        full_logits, _ = gen._logits(ids, pos, None)
        # prefill 9 then decode the 10th incrementally
        l9, caches = gen._logits(ids[:, :9], pos[:, :9], None)
        inc_logits, _ = gen._logits(ids[:, 9:], pos[:, 9:], caches)
    assert torch.allclose(full_logits, inc_logits, atol=1e-4, rtol=1e-4)


def test_greedy_deterministic():
    gpt = tiny_gpt().eval()
    gen = GPTForGeneration(gpt, {"max_dec_len": 8,
                                 "decoding_strategy": "greedy_search",
                                 "eos_token_id": 127})
    ids = torch.randint(0, 127, (2, 5))
    out1 = gen(ids)
    out2 = gen(ids)
    assert out1.shape[0] == 2 and out1.shape[1] <= 8
    assert torch.equal(out1, out2)


def test_sampling_shapes_and_eos_padding():
    torch.manual_seed(3)
    gpt = tiny_gpt().eval()
    gen = GPTForGeneration(gpt, {"max_dec_len": 6, "top_k": 8, "top_p": 0.9,
                                 "eos_token_id": 0, "pad_token_id": 0})
    out = gen(torch.randint(1, 127, (3, 4)))
    assert out.shape[0] == 3 and 1 <= out.shape[1] <= 6


def test_topk_process():
    probs = torch.tensor([[0.1, 0.5, 0.2, 0.2]])
    out = TopKProcess(probs, 2)
    assert out[0, 0] == 0 and out[0, 1] == 0.5


def test_topp_process_keeps_nucleus():
    probs = torch.tensor([[0.5, 0.3, 0.15, 0.05]])
    out = TopPProcess(probs, 0.75)
    # 0.5+0.3 = 0.8 > 0.75, first crossing token stays -> {0.5, 0.3}
    assert out[0, 0] == 0.5 and out[0, 1] == 0.3
    assert out[0, 2] == 0 and out[0, 3] == 0


def test_min_length_processor():
    proc = MinLengthLogitsProcessor(min_length=5, eos_token_id=2)
    ids = torch.zeros(1, 3, dtype=torch.long)
    logits = torch.zeros(1, 10)
    out = proc(ids, logits)
    assert out[0, 2] == -float("inf")
    ids6 = torch.zeros(1, 6, dtype=torch.long)
    assert proc(ids6, logits)[0, 2] == 0


def test_repetition_penalty():
    proc = RepetitionPenaltyLogitsProcessor(2.0)
    ids = torch.tensor([[1, 3]])
    logits = torch.tensor([[1.0, 2.0, 1.0, -2.0]])
    out = proc(ids, logits)
    assert out[0, 1] == 1.0      # positive score divided
    assert out[0, 3] == -4.0     # negative score multiplied
    assert out[0, 0] == 1.0      # untouched


def test_get_logits_processor_assembly():
    procs = get_logits_processor(min_length=3, eos_token_id=1,
                                 repetition_penalty=1.5)
    assert len(procs) == 2


def test_topp_sampling_ref_kernel():
    from paddlefleetx_amd.ops import topp_sampling
    torch.manual_seed(0)
    probs = torch.softmax(torch.randn(4, 64), dim=-1)
    ids, pp = topp_sampling(probs, torch.full((4,), 0.5), seed=7)
    assert ids.shape == (4, 1)
    # every drawn id must be inside its row nucleus
    for b in range(4):
        sorted_p, sorted_i = torch.sort(probs[b], descending=True)
        cum = torch.cumsum(sorted_p, 0)
        k = int((cum < 0.5).sum()) + 1
        assert ids[b, 0] in sorted_i[:k]


def test_tokenizer_roundtrip_synthetic_vocab():
    from paddlefleetx_amd.data.tokenizers import GPTTokenizer
    from paddlefleetx_amd.data.tokenizers.gpt_tokenizer import bytes_to_unicode
    b2u = bytes_to_unicode()
    # vocab: all 256 byte symbols + merges for "he", "ll"
    vocab = {c: i for i, c in enumerate(b2u.values())}
    he = b2u[ord("h")] + b2u[ord("e")]
    ll = b2u[ord("l")] + b2u[ord("l")]
    vocab[he] = len(vocab)
    vocab[ll] = len(vocab)
    vocab["<|endoftext|>"] = len(vocab)
    merges = [(b2u[ord("h")], b2u[ord("e")]), (b2u[ord("l")], b2u[ord("l")])]
    tok = GPTTokenizer(vocab, merges)
    ids = tok.encode("hello")
    assert tok.decode(ids) == "hello"
    assert len(ids) == 3  # he + ll + o
    padded = tok.pad([[1, 2], [3]], max_length=4)
    assert padded["input_ids"][1] == [3, tok.pad_token_id, tok.pad_token_id,
                                      tok.pad_token_id]
    assert padded["attention_mask"][0] == [1, 1, 0, 0]


def test_num_return_sequences_expansion():
    """sampling + num_return_sequences=n returns [B*n, len] grouped by
    prompt (reference expand_inputs_for_generation)."""
    torch.manual_seed(5)
    gpt = tiny_gpt().eval()
    gen = GPTForGeneration(gpt, {"max_dec_len": 5, "top_k": 8,
                                 "num_return_sequences": 3,
                                 "eos_token_id": 127, "pad_token_id": 0})
    out = gen(torch.randint(0, 127, (2, 4)))
    assert out.shape[0] == 6 and out.shape[1] <= 5
    # greedy ignores it (reference expands only the sampling branch)
    gen2 = GPTForGeneration(gpt, {"max_dec_len": 5,
                                  "decoding_strategy": "greedy_search",
                                  "num_return_sequences": 3,
                                  "eos_token_id": 127})
    assert gen2(torch.randint(0, 127, (2, 4))).shape[0] == 2
