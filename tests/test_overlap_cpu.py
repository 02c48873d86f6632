"""Overlapped DP grad allreduce == synchronous reduce (gloo, world 2)."""

import pytest
import torch
import torch.distributed as dist

from tests.test_distributed_cpu import _init, _run


def _worker(rank, world, port):
    hcg = _init(rank, world, port, dp=2)
    from paddlefleetx_amd.core.engine import EagerEngine
    from paddlefleetx_amd.models import build_module

    def make_engine(overlap):
        cfg = {
            "Global": {"global_batch_size": 4},
            "Engine": {"mix_precision": {"enable": False},
                       "accumulate_steps": 2},
            "Model": {"name": "GPTModule", "vocab_size": 128,
                      "hidden_size": 32, "num_layers": 2,
                      "num_attention_heads": 2,
                      "max_position_embeddings": 16,
                      "hidden_dropout_prob": 0.0,
                      "attention_probs_dropout_prob": 0.0,
                      "fused_attn": False},
            "Optimizer": {"name": "FusedAdamW", "weight_decay": 0.0,
                          "lr": {"name": "ConstantLR",
                                 "learning_rate": 1e-3}},
            "Distributed": {"dp_degree": 2, "reduce_overlap": overlap},
        }
        from paddlefleetx_amd.parallel.env import set_seed
        set_seed(1234)        # reset RNG tracker state between variants
        torch.manual_seed(7)  # identical init across ranks AND variants
        module = build_module(cfg)
        return EagerEngine(cfg, module)

    def run_steps(engine):
        losses = []
        for s in range(3):
            torch.manual_seed(1000 + 10 * s + rank)  # rank-specific data
            batch = (torch.randint(0, 128, (4, 16)),
                     torch.arange(16).repeat(4, 1),
                     torch.randint(0, 128, (4, 16)),
                     torch.ones(4, 16))
            losses.append(float(engine._fit_impl(batch)))
        return losses, [b.model_flat.clone()
                        for b in engine.optimizer.buckets]

    e_sync = make_engine(False)
    assert not e_sync._overlap_reduce
    l_sync, p_sync = run_steps(e_sync)

    e_ov = make_engine(True)
    assert e_ov._overlap_reduce, "overlap should engage for dp2 FusedAdamW"
    l_ov, p_ov = run_steps(e_ov)

    for a, b in zip(l_sync, l_ov):
        assert abs(a - b) < 1e-6, (a, b)
    for a, b in zip(p_sync, p_ov):
        assert torch.allclose(a, b, atol=1e-6), (a - b).abs().max()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_overlap_reduce_matches_sync():
    _run(_worker, 2)
