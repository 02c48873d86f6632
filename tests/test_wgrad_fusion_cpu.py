"""In-GEMM weight-grad accumulation (ops/linear.py): the fused path must
produce bit-identical bucket gradients and updated params vs plain
autograd, across micro-batch accumulation."""

import os

import torch

from paddlefleetx_amd.ops.linear import set_wgrad_fusion

REPO = os.path.join(os.path.dirname(__file__), "..")


def _run_engine(fused: bool):
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    import torch.distributed as dist
    if not dist.is_initialized():
        set_hcg(HybridTopology())
    set_seed(1234)  # reset the mp RNG tracker between builds
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.core import EagerEngine
    from paddlefleetx_amd.utils.config import get_config
    cfg = get_config(os.path.join(
        REPO, "paddlefleetx_amd/configs/nlp/gpt/pretrain_gpt_345M_single_card.yaml"),
        overrides=["Model.hidden_size=32", "Model.num_layers=2",
                   "Model.num_attention_heads=4", "Model.vocab_size=128",
                   "Model.max_position_embeddings=32",
                   "Model.hidden_dropout_prob=0.0",
                   "Model.attention_probs_dropout_prob=0.0",
                   "Global.micro_batch_size=2", "Global.local_batch_size=8",
                   "Engine.mix_precision.enable=False"])
    torch.manual_seed(1234)
    module = build_module(cfg)
    engine = EagerEngine(cfg, module)
    set_wgrad_fusion(fused)
    torch.manual_seed(0)
    batch = (torch.randint(0, 128, (8, 32)),
             torch.arange(32).repeat(8, 1),
             torch.randint(0, 128, (8, 32)), torch.ones(8, 32))
    # capture grads mid-step by monkeypatching past the optimizer update
    grads = None
    orig = engine.optimizer.step

    def capture(**kw):
        nonlocal grads
        grads = [b.grad_flat.clone() for b in engine.optimizer.buckets]
        return orig(**kw)

    engine.optimizer.step = capture
    loss = engine._fit_impl(batch)
    models = [b.model_flat.clone() for b in engine.optimizer.buckets]
    set_wgrad_fusion(False)
    return float(loss), grads, models


def test_wgrad_fusion_bitwise_equivalent():
    l0, g0, m0 = _run_engine(False)
    l1, g1, m1 = _run_engine(True)
    assert l0 == l1
    for a, b in zip(g0, g1):
        assert torch.allclose(a, b, atol=1e-6), (a - b).abs().max()
    for a, b in zip(m0, m1):
        assert torch.allclose(a, b, atol=1e-6), (a - b).abs().max()


def test_wgrad_fusion_with_recompute():
    """Activation recompute re-runs forward INSIDE backward: the fused
    wgrad must accumulate exactly once and match no-recompute grads."""
    import torch
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    import torch.distributed as dist
    if not dist.is_initialized():
        set_hcg(HybridTopology())
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.core import EagerEngine
    from paddlefleetx_amd.utils.config import get_config

    def run(recompute):
        set_seed(1234)
        cfg = get_config(os.path.join(
            REPO, "paddlefleetx_amd/configs/nlp/gpt/"
            "pretrain_gpt_345M_single_card.yaml"),
            overrides=["Model.hidden_size=32", "Model.num_layers=2",
                       "Model.num_attention_heads=4",
                       "Model.vocab_size=128",
                       "Model.max_position_embeddings=32",
                       "Model.hidden_dropout_prob=0.0",
                       "Model.attention_probs_dropout_prob=0.0",
                       f"Model.use_recompute={recompute}",
                       "Model.recompute_granularity=full_attn",
                       "Global.micro_batch_size=2",
                       "Global.local_batch_size=4",
                       "Engine.mix_precision.enable=False"])
        torch.manual_seed(9)
        module = build_module(cfg)
        engine = EagerEngine(cfg, module)
        set_wgrad_fusion(True)
        torch.manual_seed(1)
        batch = (torch.randint(0, 128, (4, 32)),
                 torch.arange(32).repeat(4, 1),
                 torch.randint(0, 128, (4, 32)), torch.ones(4, 32))
        grads = None
        orig = engine.optimizer.step

        def capture(**kw):
            nonlocal grads
            grads = [b.grad_flat.clone() for b in engine.optimizer.buckets]
            return orig(**kw)

        engine.optimizer.step = capture
        loss = engine._fit_impl(batch)
        set_wgrad_fusion(False)
        return float(loss), grads

    l0, g0 = run(False)
    l1, g1 = run(True)
    assert abs(l0 - l1) < 1e-6
    for a, b in zip(g0, g1):
        assert torch.allclose(a, b, atol=1e-5), (a - b).abs().max()
