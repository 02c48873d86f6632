"""World-4 parallelism compositions not covered elsewhere:
SP(mp2) × ZeRO-sharding2 and CP2 × DP2 — each runs two engine steps on
gloo and asserts the replica-consistency invariant: ranks that are data
replicas of each other (sharding peers, dp peers) must hold IDENTICAL
parameters after the optimizer step."""

import multiprocessing as mp
import os

import pytest
import torch
import torch.distributed as dist

REPO = os.path.join(os.path.dirname(__file__), "..")


def _checksums(model):
    with torch.no_grad():
        return torch.stack([p.float().sum() for p in model.parameters()])


def _assert_replicas_equal(model, group):
    """Every rank of `group` must hold the same parameter values."""
    cs = _checksums(model)
    gathered = [torch.empty_like(cs) for _ in range(group.world_size)]
    dist.all_gather(gathered, cs, group=group.group)
    for g in gathered[1:]:
        assert torch.allclose(g, gathered[0], atol=1e-5), \
            (g - gathered[0]).abs().max()


def _engine(cfg_extra, model_extra=None):
    from paddlefleetx_amd.core.engine import EagerEngine
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.parallel.env import set_seed
    set_seed(1234)
    cfg = {
        "Global": {"global_batch_size": 8},
        "Engine": {"mix_precision": {"enable": False},
                   "accumulate_steps": 1, "logging_freq": 100},
        "Model": {"name": "GPTModule", "vocab_size": 128, "hidden_size": 32,
                  "num_layers": 2, "num_attention_heads": 2,
                  "max_position_embeddings": 16,
                  "hidden_dropout_prob": 0.0,
                  "attention_probs_dropout_prob": 0.0, "fused_attn": False,
                  **(model_extra or {})},
        "Optimizer": {"name": "FusedAdamW", "weight_decay": 0.01,
                      "lr": {"name": "ConstantLR", "learning_rate": 1e-3}},
        "Distributed": cfg_extra,
    }
    module = build_module(cfg)
    return EagerEngine(cfg, module)


def _sp_shard_worker(rank, world, port):
    import sys
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.parallel.env import get_hcg, set_hcg
    from paddlefleetx_amd.parallel.topology import HybridTopology
    set_hcg(HybridTopology(mp=2, sharding=2))
    engine = _engine({"mp_degree": 2,
                      "sharding": {"sharding_degree": 2,
                                   "sharding_stage": 2}},
                     {"sequence_parallel": True})
    for s in range(2):
        g = torch.Generator().manual_seed(
            500 + s + 31 * get_hcg().get_sharding_parallel_rank())
        batch = (torch.randint(0, 128, (4, 16), generator=g),
                 torch.arange(16).repeat(4, 1),
                 torch.randint(0, 128, (4, 16), generator=g),
                 torch.ones(4, 16))
        loss = engine._fit_impl(batch)
        assert torch.isfinite(loss), s
    # sharding peers are data replicas: same params after the step
    _assert_replicas_equal(engine.module.model,
                           get_hcg().get_sharding_parallel_group())
    dist.barrier()
    dist.destroy_process_group()


def _cp_dp_worker(rank, world, port):
    import sys
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.parallel.env import get_hcg, set_hcg
    from paddlefleetx_amd.parallel.topology import HybridTopology
    set_hcg(HybridTopology(dp=2, cp=2))
    engine = _engine({"dp_degree": 2, "cp_degree": 2})
    for s in range(2):
        # cp peers must see the SAME sample stream (they shard the
        # sequence of one replica); dp peers see different streams
        g = torch.Generator().manual_seed(
            700 + s + 31 * get_hcg().get_data_parallel_rank())
        batch = (torch.randint(0, 128, (4, 16), generator=g),
                 torch.arange(16).repeat(4, 1),
                 torch.randint(0, 128, (4, 16), generator=g),
                 torch.ones(4, 16))
        batch = engine.module.pretreating_batch(batch)
        loss = engine._fit_impl(batch)
        assert torch.isfinite(loss), s
    # dp peers are replicas: allreduced grads -> identical params
    _assert_replicas_equal(engine.module.model,
                           get_hcg().get_data_parallel_group())
    dist.barrier()
    dist.destroy_process_group()


def _spawn(target, world):
    from port_util import free_port
    port = free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=target, args=(r, world, port))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(300)
        assert p.exitcode == 0, f"worker failed with {p.exitcode}"


@pytest.mark.timeout(600)
def test_sp_with_sharding_world4():
    _spawn(_sp_shard_worker, 4)


@pytest.mark.timeout(600)
def test_cp_with_dp_world4():
    _spawn(_cp_dp_worker, 4)
