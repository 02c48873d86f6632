"""End-to-end CPU slice: tiny GPT pretrain steps through the full engine."""

import os
import subprocess
import sys

import pytest
import torch

REPO = os.path.join(os.path.dirname(__file__), "..")
CFG = os.path.join(REPO, "paddlefleetx_amd", "configs", "nlp", "gpt",
                   "pretrain_gpt_345M_single_card.yaml")

TINY = [
    "Model.hidden_size=64", "Model.num_layers=2", "Model.num_attention_heads=4",
    "Model.max_position_embeddings=64", "Model.vocab_size=256",
    "Data.Train.dataset.seq_len=64", "Data.Train.dataset.vocab_size=256",
    "Data.Train.dataset.num_samples=64", "Data.Train.loader.num_workers=0",
    "Global.micro_batch_size=2", "Global.local_batch_size=4",
    "Global.max_steps=3", "Global.logging_freq=1", "Global.eval_freq=",
    "Global.save_steps=", "Engine.mix_precision.enable=False",
]


def test_train_cli_tiny(tmp_path):
    cmd = [sys.executable, os.path.join(REPO, "tools", "train.py"), "-c", CFG]
    for o in TINY + [f"Global.output_dir={tmp_path}"]:
        cmd += ["-o", o]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout + "\n" + r.stderr
    assert "ips_total" in r.stdout


def test_loss_decreases_on_overfit():
    """Tiny model overfits one repeated batch -> loss must drop."""
    from paddlefleetx_amd.utils.config import get_config
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.core import EagerEngine
    from paddlefleetx_amd.parallel.env import init_dist_env

    cfg = get_config(CFG, overrides=TINY + [
        "Optimizer.lr.name=ConstantLR", "Optimizer.lr.learning_rate=1e-3"])
    init_dist_env(cfg)
    module = build_module(cfg)
    engine = EagerEngine(cfg, module)

    torch.manual_seed(0)
    batch = (torch.randint(0, 256, (2, 64)),
             torch.arange(64).unsqueeze(0).repeat(2, 1),
             torch.randint(0, 256, (2, 64)),
             torch.ones(2, 64))
    losses = []
    for _ in range(30):
        losses.append(float(engine._fit_impl(batch)))
    assert losses[-1] < losses[0] * 0.7, losses


def test_checkpoint_roundtrip(tmp_path):
    from paddlefleetx_amd.utils.config import get_config
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.core import EagerEngine
    from paddlefleetx_amd.parallel.env import init_dist_env

    cfg = get_config(CFG, overrides=TINY + [f"Global.output_dir={tmp_path}"])
    init_dist_env(cfg)
    module = build_module(cfg)
    engine = EagerEngine(cfg, module)
    batch = (torch.randint(0, 256, (2, 64)),
             torch.arange(64).unsqueeze(0).repeat(2, 1),
             torch.randint(0, 256, (2, 64)),
             torch.ones(2, 64))
    engine._fit_impl(batch)
    engine.save(epoch=0, step=5)
    ckpt = os.path.join(str(tmp_path), "epoch_0_step_5")
    assert os.path.isdir(os.path.join(ckpt, "mp_00_sharding_00_pp_00"))

    module2 = build_module(cfg)
    engine2 = EagerEngine(cfg, module2)
    engine2.load(ckpt)
    for (n1, p1), (n2, p2) in zip(module.model.named_parameters(),
                                  module2.model.named_parameters()):
        assert n1 == n2
        assert torch.equal(p1.detach(), p2.detach()), n1
    assert engine2._load_recovery["step"] == 5


def test_recompute_granularities_equivalent():
    """full / full_attn / core_attn recompute all reproduce the
    no-recompute grads (reference granularity surface,
    hybrid_model.py:377/456/637)."""
    import torch
    from paddlefleetx_amd.models.gpt.model import (GPTForPretraining,
                                                   GPTModel,
                                                   GPTPretrainingCriterion)
    from paddlefleetx_amd.parallel.env import set_seed
    grads = {}
    for gran in (None, "full", "full_attn", "core_attn"):
        set_seed(1234)  # reset the mp-rng tracker stream between builds
        torch.manual_seed(3)
        m = GPTForPretraining(GPTModel(
            vocab_size=128, hidden_size=32, num_layers=2,
            num_attention_heads=4, max_position_embeddings=32,
            fused_attn=False, hidden_dropout_prob=0.0,
            attention_probs_dropout_prob=0.0,
            use_recompute=gran is not None,
            recompute_granularity=gran or "full"))
        m.train()
        torch.manual_seed(7)
        tokens = torch.randint(0, 128, (2, 16))
        labels = torch.randint(0, 128, (2, 16))
        loss = GPTPretrainingCriterion()(m(tokens), labels,
                                         torch.ones(2, 16))
        loss.backward()
        grads[gran] = m.gpt.layers[0].attn.qkv.weight.grad.clone()
    for gran in ("full", "full_attn", "core_attn"):
        assert torch.allclose(grads[None], grads[gran], atol=1e-6), gran


def test_recompute_replays_dropout_streams_exactly():
    """Attention dropout draws from the mp-RNG tracker, which torch's
    checkpoint does NOT preserve by itself — the checkpoint_rng_context
    snapshot/rewind (parallel/rng.py) makes the re-forward replay the
    same masks. Grads must be EXACT vs no recompute with BOTH dropouts
    active, for every granularity."""
    import torch
    from paddlefleetx_amd.models.gpt.model import (GPTForPretraining,
                                                   GPTModel,
                                                   GPTPretrainingCriterion)
    from paddlefleetx_amd.parallel.env import set_seed
    grads = {}
    for gran in (None, "full", "full_attn", "core_attn"):
        set_seed(1234)
        torch.manual_seed(3)
        m = GPTForPretraining(GPTModel(
            vocab_size=128, hidden_size=32, num_layers=2,
            num_attention_heads=4, max_position_embeddings=32,
            fused_attn=False, hidden_dropout_prob=0.1,
            attention_probs_dropout_prob=0.1,
            use_recompute=gran is not None,
            recompute_granularity=gran or "full"))
        m.train()
        torch.manual_seed(7)
        tokens = torch.randint(0, 128, (2, 16))
        labels = torch.randint(0, 128, (2, 16))
        loss = GPTPretrainingCriterion()(m(tokens), labels,
                                         torch.ones(2, 16))
        loss.backward()
        grads[gran] = m.gpt.layers[0].attn.qkv.weight.grad.clone()
    for gran in ("full", "full_attn", "core_attn"):
        assert torch.equal(grads[None], grads[gran]), gran
