"""Auto-parallel strategy planner."""

import pytest
import os

import pytest
import torch.multiprocessing as _mp

REPO = os.path.join(os.path.dirname(__file__), "..")


def _run_workers(fn, world):
    from port_util import free_port
    port = free_port()
    ctx = _mp.get_context("spawn")
    procs = [ctx.Process(target=fn, args=(r, world, port))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(200)
        assert p.exitcode == 0, f"worker failed with {p.exitcode}"


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


# ---------------------------------------------------------------------------
# shard_tensor annotation surface (reference auto_model.py:92-713)
# ---------------------------------------------------------------------------

def test_shard_tensor_annotations_single():
    import torch
    from paddlefleetx_amd.parallel.auto_shard import (ProcessMesh,
                                                      annotate_gpt,
                                                      collect_annotations,
                                                      shard_tensor,
                                                      validate_against_topology)
    from paddlefleetx_amd.parallel.env import set_hcg
    from paddlefleetx_amd.parallel.topology import HybridTopology
    import torch.distributed as dist
    if not dist.is_initialized():
        set_hcg(HybridTopology())
    mesh = ProcessMesh([1, 1, 1], ("dp", "mp", "pp"))
    p = torch.nn.Parameter(torch.randn(8, 4))
    shard_tensor(p, mesh, [None, "mp"])
    assert p._dist_attr["dims_mapping"] == (None, "mp")

    from paddlefleetx_amd.models.gpt.model import GPTModel
    m = GPTModel(vocab_size=128, hidden_size=32, num_layers=1,
                 num_attention_heads=4, max_position_embeddings=16,
                 fused_attn=False)
    n = annotate_gpt(m, mesh)
    ann = collect_annotations(m)
    assert n == len(ann) == sum(1 for _ in m.parameters())
    assert validate_against_topology(m) == []


def _mp2_annotation_worker(rank, world, port):
    import sys, os
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.parallel.topology import HybridTopology
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    set_hcg(HybridTopology(mp=2)); set_seed(7)
    from paddlefleetx_amd.parallel.auto_shard import (ProcessMesh,
                                                      annotate_gpt,
                                                      shard_tensor,
                                                      validate_against_topology)
    from paddlefleetx_amd.models.gpt.model import GPTModel
    m = GPTModel(vocab_size=128, hidden_size=32, num_layers=1,
                 num_attention_heads=4, max_position_embeddings=16,
                 fused_attn=False)
    mesh = ProcessMesh([1, 2, 1], ("dp", "mp", "pp"))
    annotate_gpt(m, mesh)
    # consistent with the eager TP sharding
    assert validate_against_topology(m) == []
    # a WRONG annotation is caught
    qkv_w = m.layers[0].attn.qkv.weight
    shard_tensor(qkv_w, mesh, [None, "mp"])  # layer shards dim 0, not 1
    problems = validate_against_topology(m)
    assert problems and "qkv" in problems[0]
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_shard_tensor_annotations_mp2():
    _run_workers(_mp2_annotation_worker, 2)
