"""GPT end-to-end under TP2 + sequence parallel (gloo world 2):
SP on must produce the same loss as SP off (same weights, same data)."""

import pytest
import torch
import torch.distributed as dist

from tests.test_distributed_cpu import _init, _run


def _worker(rank, world, port):
    hcg = _init(rank, world, port, mp_deg=2)
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.parallel.env import set_seed

    def build(sp):
        set_seed(1234)  # identical TP shard init across variants
        torch.manual_seed(11)
        cfg = {
            "Global": {"global_batch_size": 2},
            "Engine": {"mix_precision": {"enable": False}},
            "Model": {"name": "GPTModule", "vocab_size": 128,
                      "hidden_size": 32, "num_layers": 2,
                      "num_attention_heads": 2,
                      "max_position_embeddings": 16,
                      "hidden_dropout_prob": 0.0,
                      "attention_probs_dropout_prob": 0.0,
                      "fused_attn": False, "sequence_parallel": sp},
            "Distributed": {"mp_degree": 2},
        }
        return build_module(cfg)

    m_plain = build(False)
    m_sp = build(True)
    # same parallel shards -> copy weights across (state dict keys match)
    m_sp.model.load_state_dict(m_plain.model.state_dict())

    torch.manual_seed(77)
    batch = (torch.randint(0, 128, (2, 16)),
             torch.arange(16).repeat(2, 1),
             torch.randint(0, 128, (2, 16)),
             torch.ones(2, 16))
    l_plain = m_plain.training_step(batch)
    l_sp = m_sp.training_step(batch)
    assert abs(float(l_plain) - float(l_sp)) < 5e-3, \
        (float(l_plain), float(l_sp))
    l_sp.backward()
    # SP grads exist on SP-marked LayerNorm params
    for n, p in m_sp.model.named_parameters():
        if getattr(p, "sequence_parallel", False):
            assert p.grad is not None, n
            break
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_gpt_sp_matches_plain_tp():
    _run(_worker, 2)
