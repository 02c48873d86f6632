"""DP2 x TP2 x PP2 combined smoke on gloo world 8 — the driver's 8-GPU
topology, exercised end to end on CPU (tiny model, 2 steps)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

REPO = os.path.join(os.path.dirname(__file__), "..")


def _worker(rank, world, port, vpp=1):
    import sys
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.core.engine import EagerEngine
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    hcg = HybridTopology(dp=2, mp=2, pp=2)
    set_hcg(hcg)
    set_seed(1234)

    cfg = {
        "Global": {"global_batch_size": 8},
        "Engine": {"mix_precision": {"enable": False},
                   "accumulate_steps": 2},
        "Model": {"name": "GPTModule", "vocab_size": 128, "hidden_size": 32,
                  "num_layers": 4, "num_attention_heads": 2,
                  "max_position_embeddings": 16,
                  "hidden_dropout_prob": 0.0,
                  "attention_probs_dropout_prob": 0.0, "fused_attn": False},
        "Optimizer": {"name": "FusedAdamW", "weight_decay": 0.0,
                      "lr": {"name": "ConstantLR", "learning_rate": 1e-3}},
        "Distributed": {"dp_degree": 2, "mp_degree": 2, "pp_degree": 2,
                        "pipeline": {"virtual_pp_degree": vpp}},
    }
    module = build_module(cfg)
    engine = EagerEngine(cfg, module)
    losses = []
    for s in range(2):
        torch.manual_seed(500 + s + 31 * hcg.get_data_parallel_rank())
        batch = (torch.randint(0, 128, (4, 16)),
                 torch.arange(16).repeat(4, 1),
                 torch.randint(0, 128, (4, 16)),
                 torch.ones(4, 16))
        loss = engine._fit_impl(batch)
        losses.append(float(loss))
    # last pp stage computes the loss; all ranks must stay in sync
    if hcg.is_last_stage():
        assert all(l > 0 and l < 20 for l in losses), losses
    # every rank reaches the barrier -> schedule is deadlock-free
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_dp2_tp2_pp2_world8():
    from port_util import free_port; port = free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, 8, port))
             for r in range(8)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(500)
        assert p.exitcode == 0, f"worker failed with {p.exitcode}"


@pytest.mark.timeout(600)
def test_dp2_tp2_pp2_vpp2_world8():
    """The driver topology with interleaved virtual stages on top
    (4 layers -> 4 model chunks, acc=2 == pp_degree)."""
    from port_util import free_port; port = free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, 8, port, 2))
             for r in range(8)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(500)
        assert p.exitcode == 0, f"worker failed with {p.exitcode}"


def _ckpt_worker(rank, world, port, tmpdir, vpp=1):
    import sys
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.core.engine import EagerEngine
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    hcg = HybridTopology(pp=2)
    set_hcg(hcg)
    set_seed(1234)
    cfg = {
        "Global": {"global_batch_size": 4},
        "Engine": {"mix_precision": {"enable": False},
                   "accumulate_steps": 2,
                   "save_load": {"output_dir": tmpdir}},
        "Model": {"name": "GPTModule", "vocab_size": 128, "hidden_size": 32,
                  "num_layers": 4, "num_attention_heads": 2,
                  "max_position_embeddings": 16,
                  "hidden_dropout_prob": 0.0,
                  "attention_probs_dropout_prob": 0.0, "fused_attn": False},
        "Optimizer": {"name": "FusedAdamW", "weight_decay": 0.0,
                      "lr": {"name": "ConstantLR", "learning_rate": 1e-3}},
        "Distributed": {"pp_degree": 2,
                        "pipeline": {"virtual_pp_degree": vpp}},
    }
    module = build_module(cfg)
    engine = EagerEngine(cfg, module)
    torch.manual_seed(77)
    batch = (torch.randint(0, 128, (4, 16)),
             torch.arange(16).repeat(4, 1),
             torch.randint(0, 128, (4, 16)), torch.ones(4, 16))
    engine._fit_impl(batch)
    engine.save(epoch=0, step=7)
    dist.barrier()
    ckpt = os.path.join(tmpdir, "epoch_0_step_7")
    # per-(mp, sharding, pp) shard layout (reference io.py:56-58)
    mydir = os.path.join(ckpt, f"mp_00_sharding_00_pp_{hcg.pp_rank:02d}")
    assert os.path.isdir(mydir), mydir

    module2 = build_module(cfg)
    engine2 = EagerEngine(cfg, module2)
    engine2.load(ckpt)
    for (n1, p1), (n2, p2) in zip(module.model.named_parameters(),
                                  module2.model.named_parameters()):
        assert n1 == n2
        assert torch.equal(p1.detach(), p2.detach()), n1
    assert engine2._load_recovery["step"] == 7
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_pp2_checkpoint_roundtrip():
    import tempfile
    from port_util import free_port
    port = free_port()
    ctx = mp.get_context("spawn")
    with tempfile.TemporaryDirectory() as tmpdir:
        procs = [ctx.Process(target=_ckpt_worker, args=(r, 2, port, tmpdir))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(300)
            assert p.exitcode == 0, f"worker failed with {p.exitcode}"


def _fp16_pp_worker(rank, world, port):
    """fp16 dynamic loss scaling composed with the 1F1B pipeline: the
    scale rides `forward_backward_pipeline(scale=...)`, grads unscale in
    the optimizer step, and the loss/params stay finite."""
    import sys
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.core.engine import EagerEngine
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    hcg = HybridTopology(pp=2)
    set_hcg(hcg)
    set_seed(1234)
    cfg = {
        "Global": {"global_batch_size": 4},
        "Engine": {"mix_precision": {"enable": True, "dtype": "float16",
                                     "scale_loss": 1024.0},
                   "accumulate_steps": 2},
        "Model": {"name": "GPTModule", "vocab_size": 128, "hidden_size": 32,
                  "num_layers": 4, "num_attention_heads": 2,
                  "max_position_embeddings": 16,
                  "hidden_dropout_prob": 0.0,
                  "attention_probs_dropout_prob": 0.0, "fused_attn": False},
        "Optimizer": {"name": "FusedAdamW", "weight_decay": 0.0,
                      "lr": {"name": "ConstantLR", "learning_rate": 1e-3}},
        "Distributed": {"pp_degree": 2},
    }
    module = build_module(cfg)
    engine = EagerEngine(cfg, module)
    assert engine.loss_scale == 1024.0
    torch.manual_seed(31)
    batch = (torch.randint(0, 128, (4, 16)),
             torch.arange(16).repeat(4, 1),
             torch.randint(0, 128, (4, 16)), torch.ones(4, 16))
    for _ in range(2):
        loss = engine._fit_impl(batch)
    if hcg.is_last_stage():
        assert torch.isfinite(loss), loss
    for p in module.model.parameters():
        assert torch.isfinite(p).all()
    assert engine._found_inf == 0.0
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_fp16_scaler_with_pipeline():
    from port_util import free_port
    port = free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_fp16_pp_worker, args=(r, 2, port))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(300)
        assert p.exitcode == 0, f"worker failed with {p.exitcode}"


@pytest.mark.timeout(600)
def test_pp2_vpp2_checkpoint_roundtrip():
    """Interleaved virtual stages save/load through the same shard-dir
    layout (each rank holds two non-adjacent chunks)."""
    import tempfile
    from port_util import free_port
    port = free_port()
    ctx = mp.get_context("spawn")
    with tempfile.TemporaryDirectory() as tmpdir:
        procs = [ctx.Process(target=_ckpt_worker, args=(r, 2, port, tmpdir, 2))
                 for r in range(2)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(300)
            assert p.exitcode == 0, f"worker failed with {p.exitcode}"
