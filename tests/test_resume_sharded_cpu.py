"""Checkpoint-resume equivalence under ZeRO sharding (gloo world 2,
sharding_degree 2): each rank saves its own optimizer shard
(mp_00_sharding_0X_pp_00 layout), a fresh engine reloads it, skips the
consumed batches and finishes — final weights match the continuous
run."""

import multiprocessing as mp
import os
import shutil
import tempfile

import pytest
import torch
import torch.distributed as dist

REPO = os.path.join(os.path.dirname(__file__), "..")


def _cfg(outdir, extra_engine=None):
    cfg = {
        "Global": {"global_batch_size": 8},
        "Engine": {"mix_precision": {"enable": False},
                   "accumulate_steps": 1, "logging_freq": 100,
                   "save_load": {"output_dir": outdir}},
        "Model": {"name": "GPTModule", "vocab_size": 128, "hidden_size": 32,
                  "num_layers": 2, "num_attention_heads": 2,
                  "max_position_embeddings": 16,
                  "hidden_dropout_prob": 0.0,
                  "attention_probs_dropout_prob": 0.0, "fused_attn": False},
        "Optimizer": {"name": "FusedAdamW", "weight_decay": 0.01,
                      "lr": {"name": "ConstantLR", "learning_rate": 1e-3}},
        "Distributed": {"sharding": {"sharding_degree": 2,
                                     "sharding_stage": 1}},
    }
    if extra_engine:
        cfg["Engine"].update(extra_engine)
    return cfg


def _worker(rank, world, port, outdir):
    import sys
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.core import EagerEngine
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    set_hcg(HybridTopology(sharding=2))

    def build(extra):
        set_seed(1234)
        cfg = _cfg(outdir, extra)
        module = build_module(cfg)
        return EagerEngine(cfg, module)

    def batches(n):
        # sharding ranks are data-parallel replicas of the data stream:
        # each rank has its own deterministic shard-stream
        g = torch.Generator().manual_seed(1000 + rank)
        out = []
        for _ in range(n):
            out.append((torch.randint(0, 128, (4, 16), generator=g),
                        torch.arange(16).repeat(4, 1),
                        torch.randint(0, 128, (4, 16), generator=g),
                        torch.ones(4, 16)))
        return out

    a = build({"max_steps": 4})
    a.fit(batches(4))
    ref = {k: v.clone() for k, v in a.module.model.state_dict().items()}

    b = build({"max_steps": 2})
    b.fit(batches(4))
    b.save(0, 2)
    dist.barrier()  # both sharding ranks must finish writing their shard
    ckpt = os.path.join(outdir, "epoch_0_step_2")
    assert os.path.isdir(os.path.join(
        ckpt, f"mp_00_sharding_{rank:02d}_pp_00"))

    c = build({"max_steps": 4, "save_load": {"output_dir": outdir,
                                             "ckpt_dir": ckpt}})
    assert c._load_recovery["step"] == 2
    c.fit(batches(4))
    got = c.module.model.state_dict()
    for k, v in ref.items():
        assert torch.allclose(got[k].float(), v.float(), atol=1e-6), k

    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_sharded_resume_matches_continuous():
    from port_util import free_port
    port = free_port()
    outdir = tempfile.mkdtemp(prefix="resume_shard_")
    try:
        ctx = mp.get_context("spawn")
        procs = [ctx.Process(target=_worker, args=(r, 2, port, outdir))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(240)
            assert p.exitcode == 0, f"worker failed with {p.exitcode}"
    finally:
        shutil.rmtree(outdir, ignore_errors=True)
