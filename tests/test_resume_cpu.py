"""Checkpoint-resume equivalence: save at step k, rebuild with
`Engine.save_load.ckpt_dir`, resume (skip_until k), and finish — final
weights must equal a continuous run over the same batch stream
(reference eager_engine save/load + consumed-samples resume)."""

import os
import tempfile

import torch

from paddlefleetx_amd.parallel.env import set_hcg, set_seed
from paddlefleetx_amd.parallel.topology import HybridTopology


def _cfg(extra_engine=None):
    cfg = {
        "Global": {"global_batch_size": 4},
        "Engine": {"mix_precision": {"enable": False},
                   "accumulate_steps": 1, "logging_freq": 100},
        "Model": {"name": "GPTModule", "vocab_size": 128, "hidden_size": 32,
                  "num_layers": 2, "num_attention_heads": 2,
                  "max_position_embeddings": 16,
                  # dropout ON: resume must also restore the global AND
                  # mp-tracker RNG streams to match the continuous run
                  "hidden_dropout_prob": 0.1,
                  "attention_probs_dropout_prob": 0.1,
                  "fused_attn": False},
        "Optimizer": {"name": "FusedAdamW", "weight_decay": 0.01,
                      "lr": {"name": "ConstantLR", "learning_rate": 1e-3}},
        "Distributed": {},
    }
    if extra_engine:
        cfg["Engine"].update(extra_engine)
    return cfg


def _build(cfg):
    from paddlefleetx_amd.core import EagerEngine
    from paddlefleetx_amd.models import build_module
    set_seed(1234)
    module = build_module(cfg)
    return EagerEngine(cfg, module)


def _batches(n):
    g = torch.Generator().manual_seed(99)
    out = []
    for _ in range(n):
        out.append((torch.randint(0, 128, (4, 16), generator=g),
                    torch.arange(16).repeat(4, 1),
                    torch.randint(0, 128, (4, 16), generator=g),
                    torch.ones(4, 16)))
    return out


def test_resume_matches_continuous_run():
    set_hcg(HybridTopology())
    batches = _batches(4)

    # continuous: 4 steps
    a = _build(_cfg({"max_steps": 4}))
    a.fit(batches)
    ref = {k: v.clone() for k, v in a.module.model.state_dict().items()}
    assert a.module.global_step == 4

    with tempfile.TemporaryDirectory() as td:
        # first half: 2 steps, then checkpoint
        b = _build(_cfg({"max_steps": 2,
                         "save_load": {"output_dir": td}}))
        b.fit(batches)
        b.save(0, 2)
        ckpt = os.path.join(td, "epoch_0_step_2")
        assert os.path.isdir(ckpt)

        # resume: loads ckpt, skips the 2 consumed batches, runs 2 more
        c = _build(_cfg({"max_steps": 4,
                         "save_load": {"output_dir": td,
                                       "ckpt_dir": ckpt}}))
        assert c._load_recovery["step"] == 2
        assert c.module.global_step == 2
        c.fit(batches)
        assert c.module.global_step == 4

    got = c.module.model.state_dict()
    for k, v in ref.items():
        assert torch.allclose(got[k].float(), v.float(), atol=1e-6), k


def test_resume_restores_loss_scale():
    """Dynamic fp16 loss scale survives save/load (a reset to the 32768
    default would burn found_inf skip steps after resume)."""
    set_hcg(HybridTopology())
    cfg = _cfg({"max_steps": 2,
                "mix_precision": {"enable": True, "dtype": "float16",
                                  "scale_loss": 4096.0}})
    with tempfile.TemporaryDirectory() as td:
        cfg["Engine"]["save_load"] = {"output_dir": td}
        a = _build(cfg)
        a.loss_scale = 512.0   # pretend the dynamic scaler backed off
        a._good_steps = 7
        a.fit(_batches(2))
        a.save(0, 2)

        cfg2 = _cfg({"max_steps": 4,
                     "mix_precision": {"enable": True, "dtype": "float16",
                                       "scale_loss": 4096.0},
                     "save_load": {"output_dir": td,
                                   "ckpt_dir": os.path.join(
                                       td, "epoch_0_step_2")}})
        b = _build(cfg2)
        assert b.loss_scale == a.loss_scale
        assert b._good_steps == a._good_steps
