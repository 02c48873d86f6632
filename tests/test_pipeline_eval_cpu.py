"""Engine evaluation under pipeline parallelism must run the
forward-only pipeline schedule (reference eager_engine.py:655
`eval_batch`), not the stage-local validation_step: a pp stage only
holds its own layers, so the stage-local path would crash or log a
partial loss. Checks the engine routes to eval_pipeline and that the
eval loss equals the training-schedule forward loss at the same
weights (pp2, gloo world 2)."""

import multiprocessing as mp
import os

import pytest
import torch
import torch.distributed as dist

REPO = os.path.join(os.path.dirname(__file__), "..")


def _worker(rank, world, port):
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
    assert engine.is_pipeline

    torch.manual_seed(7)
    batch = (torch.randint(0, 128, (4, 16)),
             torch.arange(16).repeat(4, 1),
             torch.randint(0, 128, (4, 16)),
             torch.ones(4, 16))

    # training-schedule forward loss at the initial weights (backward
    # runs but no optimizer step, so weights are unchanged)
    loss_train = float(module.model.forward_backward_pipeline(
        batch, module.loss_fn, engine.accumulate_steps))
    module.model.zero_grad(set_to_none=True)

    # engine evaluation path on the same batch must match it exactly
    logged = []
    module.validation_step_end = lambda d: logged.append(d["loss"])
    engine._evaluate_impl(0, [batch])
    assert len(logged) == 1
    assert logged[0] == pytest.approx(loss_train, abs=1e-5), \
        (logged[0], loss_train)

    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_pipeline_engine_evaluate():
    from port_util import free_port
    port = free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_worker, args=(r, 2, port))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0, f"worker failed with {p.exitcode}"


def _recompute_worker(rank, world, port):
    import sys
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.models import build_module
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    set_hcg(HybridTopology(pp=2))

    def build(rec):
        set_seed(1234)
        cfg = {
            "Global": {"global_batch_size": 4},
            "Engine": {"mix_precision": {"enable": False},
                       "accumulate_steps": 2},
            "Model": {"name": "GPTModule", "vocab_size": 128,
                      "hidden_size": 32, "num_layers": 4,
                      "num_attention_heads": 2,
                      "max_position_embeddings": 16,
                      "hidden_dropout_prob": 0.1,
                      "attention_probs_dropout_prob": 0.1,
                      "fused_attn": False, "use_recompute": rec,
                      "recompute_granularity": "full"},
            "Distributed": {"pp_degree": 2},
        }
        return build_module(cfg)

    torch.manual_seed(7)
    batch = (torch.randint(0, 128, (4, 16)),
             torch.arange(16).repeat(4, 1),
             torch.randint(0, 128, (4, 16)), torch.ones(4, 16))
    losses = {}
    for rec in (False, True):
        m = build(rec)
        torch.manual_seed(21)  # dropout stream identical across variants
        from paddlefleetx_amd.parallel.env import set_seed as _ss
        _ss(1234)
        losses[rec] = float(m.model.forward_backward_pipeline(
            batch, m.loss_fn, 2))
    # pipe-layer "full" recompute replays dropout exactly -> same loss
    assert abs(losses[False] - losses[True]) < 1e-6, losses
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_pipeline_full_recompute_with_dropout():
    from port_util import free_port
    port = free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_recompute_worker, args=(r, 2, port))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(240)
        assert p.exitcode == 0, f"worker failed with {p.exitcode}"
