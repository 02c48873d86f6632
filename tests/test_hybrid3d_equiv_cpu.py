"""DP2 x TP2 x PP2 world-8 EXACT equivalence vs a single-process run —
the driver's 8-GPU topology verified to numerical identity (loss + grads),
not just smoke. Also the composed case:
mp2 x pp2 x vpp2 + sequence parallel + loss-scaling, world 4.

Pattern follows tests/test_distributed_cpu.py:230 (pp2-vs-single): the
world-1 reference saves full per-layer state dicts; parallel workers load
them, slicing TP-sharded params by their `partition_dim` tag
(parallel/tp.py:122-159) — the fused-QKV layout is head-major [H, 3, D]
so contiguous row slicing IS head slicing.
"""

import os
import tempfile

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

REPO = os.path.join(os.path.dirname(__file__), "..")

MODEL_KW = dict(vocab_size=128, hidden_size=32, num_layers=4,
                num_attention_heads=4, max_position_embeddings=32,
                hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
                fused_attn=False, dtype=torch.float32)
SEQ = 32
GLOBAL_B = 8


def _init(rank, world, port, dp=1, mp_deg=1, pp=1):
    import sys
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.parallel.topology import HybridTopology
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    hcg = HybridTopology(dp=dp, mp=mp_deg, pp=pp)
    set_hcg(hcg)
    set_seed(1234)
    return hcg


def _global_batch():
    torch.manual_seed(5)
    return (torch.randint(0, 128, (GLOBAL_B, SEQ)),
            torch.arange(SEQ).unsqueeze(0).repeat(GLOBAL_B, 1),
            torch.randint(0, 128, (GLOBAL_B, SEQ)),
            torch.ones(GLOBAL_B, SEQ))


def _make_model(**extra):
    from paddlefleetx_amd.models.gpt.pipeline_model import GPTForPretrainingPipe
    kw = dict(MODEL_KW)
    kw.update(extra)
    return GPTForPretrainingPipe(**kw)


def _shard_of(full, param, mp_rank, mp_size):
    if mp_size > 1 and getattr(param, "is_mp", False):
        d = param.partition_dim
        n = full.shape[d] // mp_size
        return full.narrow(d, mp_rank * n, n)
    return full


def _ref_worker(rank, world, port, tmpdir, **model_extra):
    """world=1 reference: full model, full global batch, acc=4."""
    _init(rank, world, port)
    from paddlefleetx_amd.models.gpt.model import GPTPretrainingCriterion
    m = _make_model()
    for i, layer in enumerate(m.layers):
        torch.save(layer.state_dict(), os.path.join(tmpdir, f"layer_{i}.pt"))
    loss = m.forward_backward_pipeline(_global_batch(),
                                       GPTPretrainingCriterion(),
                                       accumulate_steps=4)
    grads = [{n: p.grad.clone() for n, p in layer.named_parameters()
              if p.grad is not None} for layer in m.layers]
    torch.save({"loss": loss, "grads": grads},
               os.path.join(tmpdir, "ref.pt"))
    dist.destroy_process_group()


def _load_sharded(m, tmpdir, hcg):
    mp_rank = hcg.get_model_parallel_rank()
    mp_size = hcg.get_model_parallel_world_size()
    for i, layer in enumerate(m.layers):
        gi = m._layer_desc_idx[i]
        sd = torch.load(os.path.join(tmpdir, f"layer_{gi}.pt"),
                        weights_only=False)
        params = dict(layer.named_parameters())
        with torch.no_grad():
            for name, p in params.items():
                p.copy_(_shard_of(sd[name], p, mp_rank, mp_size))
            for name, b in layer.named_buffers():
                if name in sd:
                    b.copy_(sd[name])


def _hybrid_worker(rank, world, port, tmpdir):
    """dp2 x mp2 x pp2: half the batch per dp replica, acc=2, manual DP
    grad averaging; loss and every grad must match the reference."""
    hcg = _init(rank, world, port, dp=2, mp_deg=2, pp=2)
    from paddlefleetx_amd.models.gpt.model import GPTPretrainingCriterion
    m = _make_model()
    _load_sharded(m, tmpdir, hcg)

    tokens, pos, labels, mask = _global_batch()
    dp_rank = hcg.get_data_parallel_rank()
    half = GLOBAL_B // 2
    sl = slice(dp_rank * half, (dp_rank + 1) * half)
    batch = (tokens[sl], pos[sl], labels[sl], mask[sl])
    loss = m.forward_backward_pipeline(batch, GPTPretrainingCriterion(),
                                       accumulate_steps=2)

    # DP grad average (the engine's bucketed allreduce, spelled out)
    dp_group = hcg.get_data_parallel_group().group
    for p in m.parameters():
        if p.grad is not None:
            dist.all_reduce(p.grad, group=dp_group)
            p.grad /= 2.0

    # dp-average the (pp-broadcast) loss
    loss = loss.clone()
    dist.all_reduce(loss, group=dp_group)
    loss /= 2.0

    ref = torch.load(os.path.join(tmpdir, "ref.pt"), weights_only=False)
    assert torch.allclose(loss, ref["loss"], atol=1e-5), (loss, ref["loss"])

    mp_rank = hcg.get_model_parallel_rank()
    checked = 0
    for i, layer in enumerate(m.layers):
        gi = m._layer_desc_idx[i]
        for name, p in layer.named_parameters():
            if p.grad is None:
                continue
            rg = _shard_of(ref["grads"][gi][name], p, mp_rank, 2)
            assert torch.allclose(p.grad, rg, atol=1e-4), \
                (gi, name, (p.grad - rg).abs().max())
            checked += 1
    assert checked > 0
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_dp2_tp2_pp2_world8_exact_equivalence():
    from port_util import free_port
    ctx = mp.get_context("spawn")
    with tempfile.TemporaryDirectory() as tmpdir:
        port = free_port()
        p = ctx.Process(target=_ref_worker, args=(0, 1, port, tmpdir))
        p.start(); p.join(300)
        assert p.exitcode == 0
        port = free_port()
        procs = [ctx.Process(target=_hybrid_worker,
                             args=(r, 8, port, tmpdir)) for r in range(8)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(500)
            assert p.exitcode == 0, f"worker failed with {p.exitcode}"


# ---------------------------------------------------------------------------
# Composed: mp2 x pp2 x vpp2 + sequence parallel + loss scale, world 4
# ---------------------------------------------------------------------------

def _sp_ref_worker(rank, world, port, tmpdir):
    _ref_worker(rank, world, port, tmpdir)


def _sp_vpp_worker(rank, world, port, tmpdir):
    """mp2 x pp2 with virtual_pp_degree=2 + Megatron-SP + scaled loss
    (the fp16 GradScaler path run at fp32: scale*unscale must be exact
    up to rounding)."""
    hcg = _init(rank, world, port, mp_deg=2, pp=2)
    from paddlefleetx_amd.models.gpt.model import GPTPretrainingCriterion
    scale = 1024.0
    m = _make_model(sequence_parallel=True, virtual_pp_degree=2)
    _load_sharded(m, tmpdir, hcg)
    loss = m.forward_backward_pipeline(_global_batch(),
                                       GPTPretrainingCriterion(),
                                       accumulate_steps=4, scale=scale)
    ref = torch.load(os.path.join(tmpdir, "ref.pt"), weights_only=False)
    assert torch.allclose(loss, ref["loss"], atol=1e-5), (loss, ref["loss"])

    # unscale (what MixPrecisionScaler does before the step), then
    # SP LayerNorm-param grads were reduce-scattered contributions:
    # mark_as_sp_param grads need the mp allreduce the engine performs
    from paddlefleetx_amd.parallel import sp as sp_ops
    mp_group = hcg.get_model_parallel_group().group
    mp_rank = hcg.get_model_parallel_rank()
    checked = 0
    for i, layer in enumerate(m.layers):
        gi = m._layer_desc_idx[i]
        for name, p in layer.named_parameters():
            if p.grad is None:
                continue
            g = p.grad / scale
            if getattr(p, "sequence_parallel", False) or \
                    getattr(p, "_sp_param", False):
                dist.all_reduce(g, group=mp_group)
            rg = _shard_of(ref["grads"][gi][name], p, mp_rank, 2)
            assert torch.allclose(g, rg, atol=2e-4), \
                (gi, name, (g - rg).abs().max())
            checked += 1
    assert checked > 0
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_vpp2_sp_scaled_world4_exact_equivalence():
    from port_util import free_port
    ctx = mp.get_context("spawn")
    with tempfile.TemporaryDirectory() as tmpdir:
        port = free_port()
        p = ctx.Process(target=_sp_ref_worker, args=(0, 1, port, tmpdir))
        p.start(); p.join(300)
        assert p.exitcode == 0
        port = free_port()
        procs = [ctx.Process(target=_sp_vpp_worker,
                             args=(r, 4, port, tmpdir)) for r in range(4)]
        for p in procs:
            p.start()
        for p in procs:
            p.join(500)
            assert p.exitcode == 0, f"worker failed with {p.exitcode}"
