"""Auto-parallel strategy planner."""

import pytest

from paddlefleetx_amd.core.auto_engine import (estimate_param_count,
                                               plan_strategy)


def test_param_estimates():
    m345 = dict(hidden_size=1024, num_layers=24, vocab_size=50304,
                max_position_embeddings=1024)
    m67 = dict(hidden_size=4096, num_layers=32, vocab_size=50304,
               max_position_embeddings=1024)
    m175 = dict(hidden_size=12288, num_layers=96, vocab_size=51200,
                max_position_embeddings=2048)
    assert 0.3e9 < estimate_param_count(m345) < 0.45e9
    assert 6e9 < estimate_param_count(m67) < 7.5e9
    assert 1.6e11 < estimate_param_count(m175) < 1.95e11


def test_plan_small_model_pure_dp():
    plan = plan_strategy(dict(hidden_size=1024, num_layers=24,
                              vocab_size=50304,
                              max_position_embeddings=1024), 8,
                         micro_batch=8)
    # 345M fits everywhere -> pure DP8
    assert plan["dp_degree"] == 8 and plan["mp_degree"] == 1
    assert plan["pp_degree"] == 1


def test_plan_67b_single_gpu_fits():
    # 6.7B on ONE MI355X (288 GB): params 13.4 + opt 80 GB fits
    plan = plan_strategy(dict(hidden_size=4096, num_layers=32,
                              vocab_size=50304,
                              max_position_embeddings=1024), 1,
                         micro_batch=8)
    assert plan["mp_degree"] == 1 and plan["pp_degree"] == 1


def test_plan_175b_needs_model_sharding():
    plan = plan_strategy(dict(hidden_size=12288, num_layers=96,
                              vocab_size=51200,
                              max_position_embeddings=2048,
                              use_recompute=True), 8, micro_batch=1)
    # 175B: bf16 params alone 350 GB -> must split params across GPUs
    assert plan["mp_degree"] * plan["pp_degree"] * \
        (plan["sharding_degree"] if plan["sharding_stage"] >= 3 else 1) >= 4


def test_plan_rejects_impossible():
    with pytest.raises(ValueError):
        plan_strategy(dict(hidden_size=50000, num_layers=400,
                           vocab_size=50304,
                           max_position_embeddings=2048), 1)
