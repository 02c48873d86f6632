"""T5 encoder/decoder."""

import pytest
import torch

from paddlefleetx_amd.parallel.env import set_hcg
from paddlefleetx_amd.parallel.topology import HybridTopology


@pytest.fixture(autouse=True)
def _env():
    set_hcg(HybridTopology())
    yield


def test_relative_position_buckets():
    from paddlefleetx_amd.models.t5 import relative_position_bucket
    rel = torch.arange(-8, 9).unsqueeze(0)
    b_bi = relative_position_bucket(rel, True, 32, 128)
    assert int(b_bi.min()) >= 0 and int(b_bi.max()) < 32
    # zero distance -> bucket 0
    assert int(b_bi[0, 8]) == 0
    b_causal = relative_position_bucket(rel, False, 32, 128)
    # future positions (relative > 0) clamp to bucket 0 in causal mode
    assert int(b_causal[0, -1]) == 0


def test_t5_encoder_forward_backward():
    from paddlefleetx_amd.models.t5 import T5EncoderModel
    torch.manual_seed(0)
    m = T5EncoderModel(vocab_size=100, d_model=32, d_kv=8, d_ff=64,
                       num_layers=2, num_heads=4, dropout_rate=0.0)
    ids = torch.randint(0, 100, (2, 12))
    mask = torch.ones(2, 12)
    mask[:, 8:] = 0
    out = m(ids, attention_mask=mask)
    assert out.shape == (2, 12, 32)
    out.sum().backward()
    assert m.shared.weight.grad is not None
    # padding positions must not influence valid ones
    ids2 = ids.clone()
    ids2[:, 8:] = 7
    m.zero_grad()
    out2 = m(ids2, attention_mask=mask)
    assert torch.allclose(out[:, :8], out2[:, :8], atol=1e-5)


def test_t5_encdec_forward_and_causality():
    from paddlefleetx_amd.models.t5 import T5Model
    torch.manual_seed(1)
    m = T5Model(vocab_size=64, d_model=32, d_kv=8, d_ff=64, num_layers=1,
                num_heads=4, dropout_rate=0.0).eval()
    src = torch.randint(0, 64, (1, 6))
    tgt = torch.randint(0, 64, (1, 5))
    logits = m(src, tgt)
    assert logits.shape == (1, 5, 64)
    # decoder causality: changing a later target token leaves earlier
    # positions' logits unchanged
    tgt2 = tgt.clone()
    tgt2[0, 4] = (tgt2[0, 4] + 1) % 64
    logits2 = m(src, tgt2)
    assert torch.allclose(logits[0, :4], logits2[0, :4], atol=1e-5)
    assert not torch.allclose(logits[0, 4], logits2[0, 4], atol=1e-5)


def test_t5_gated_act_variant():
    from paddlefleetx_amd.models.t5 import (T5Config, T5DenseGatedActDense,
                                            T5LayerFF)
    cfg = T5Config(d_model=16, d_ff=32, feed_forward_proj="gated-gelu",
                   dropout_rate=0.0)
    ff = T5LayerFF(cfg)
    assert isinstance(ff.DenseReluDense, T5DenseGatedActDense)
    y = ff(torch.randn(2, 4, 16))
    assert y.shape == (2, 4, 16)


def test_t5_conditional_generation():
    import torch
    from paddlefleetx_amd.models.t5 import T5ForConditionalGeneration
    torch.manual_seed(0)
    m = T5ForConditionalGeneration(vocab_size=64, d_model=32, d_ff=64,
                                   num_layers=2, num_heads=2)
    src = torch.randint(0, 64, (2, 10))
    labels = torch.randint(0, 64, (2, 6))
    loss, logits = m(src, labels=labels)
    assert torch.isfinite(loss) and logits.shape == (2, 6, 64)
    loss.backward()
    out = m.generate(src, max_length=5)
    assert out.shape[0] == 2 and out.shape[1] <= 6
    # teacher forcing path matches manual shift
    dec_in = m._shift_right(labels)
    assert dec_in[0, 0] == m.decoder_start_token_id
    assert torch.equal(dec_in[:, 1:], labels[:, :-1])
