"""Ring attention (CP backend) fwd+bwd vs full causal attention, gloo.

MI355X-native long-context extension beyond the reference (SURVEY §5:
reference has no ring attention); must reproduce the exact full-attention
forward AND gradients on sequence shards.
"""

import math
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

REPO = os.path.join(os.path.dirname(__file__), "..")


def _init(rank, world, port):
    import sys
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    set_hcg(HybridTopology(cp=world))
    set_seed(1234)


def _run(fn, world, args=()):
    from port_util import free_port; port = free_port()
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=fn, args=(r, world, port) + args)
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(300)
        assert p.exitcode == 0, f"worker failed with {p.exitcode}"


def _full_ref(q, k, v, scale):
    s = torch.matmul(q, k.transpose(-1, -2)) * scale
    S = s.shape[-1]
    mask = torch.ones(S, S, dtype=torch.bool).tril()
    s = s.masked_fill(~mask, float("-inf"))
    return torch.matmul(torch.softmax(s.float(), -1).to(q.dtype), v)


def _ring_worker(rank, world, port):
    _init(rank, world, port)
    from paddlefleetx_amd.parallel.ring import ring_attention
    torch.manual_seed(1)
    B, H, S, D = 2, 4, 32, 8
    scale = 1.0 / math.sqrt(D)
    q = torch.randn(B, H, S, D)
    k = torch.randn(B, H, S, D)
    v = torch.randn(B, H, S, D)
    do = torch.randn(B, H, S, D)

    # reference on the full sequence with autograd grads
    qr = q.clone().requires_grad_(True)
    kr = k.clone().requires_grad_(True)
    vr = v.clone().requires_grad_(True)
    ref = _full_ref(qr, kr, vr, scale)
    ref.backward(do)

    Sl = S // world
    sl = slice(rank * Sl, (rank + 1) * Sl)
    ql = q[:, :, sl].clone().requires_grad_(True)
    kl = k[:, :, sl].clone().requires_grad_(True)
    vl = v[:, :, sl].clone().requires_grad_(True)
    o = ring_attention(ql, kl, vl, scale)
    assert torch.allclose(o, ref.detach()[:, :, sl], atol=1e-5), \
        (o - ref.detach()[:, :, sl]).abs().max()
    o.backward(do[:, :, sl])
    for name, got, want in (("dq", ql.grad, qr.grad[:, :, sl]),
                            ("dk", kl.grad, kr.grad[:, :, sl]),
                            ("dv", vl.grad, vr.grad[:, :, sl])):
        assert torch.allclose(got, want, atol=1e-4), \
            (name, (got - want).abs().max())
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ring_attention_matches_full_cp2():
    _run(_ring_worker, 2)


@pytest.mark.timeout(300)
def test_ring_attention_matches_full_cp4():
    _run(_ring_worker, 4)


def _gpt_ring_worker(rank, world, port):
    """GPT training step with cp_backend=ring agrees across ranks."""
    _init(rank, world, port)
    from paddlefleetx_amd.models import build_module
    cfg = {
        "Global": {"global_batch_size": 2},
        "Engine": {"mix_precision": {"enable": False}},
        "Model": {"name": "GPTModule", "vocab_size": 128, "hidden_size": 32,
                  "num_layers": 2, "num_attention_heads": 4,
                  "max_position_embeddings": 32,
                  "hidden_dropout_prob": 0.0,
                  "attention_probs_dropout_prob": 0.0, "fused_attn": True,
                  "cp_backend": "ring"},
        "Distributed": {"cp_degree": world},
    }
    torch.manual_seed(5)
    mod = build_module(cfg)
    torch.manual_seed(9)
    tokens = torch.randint(0, 128, (2, 32))
    pos = torch.arange(32).unsqueeze(0).repeat(2, 1)
    labels = torch.randint(0, 128, (2, 32))
    mask = torch.ones(2, 32)
    batch = mod.pretreating_batch((tokens, pos, labels, mask))
    loss = mod.training_step(batch)
    loss.backward()
    t = loss.detach().clone()
    dist.broadcast(t, src=0)
    assert torch.allclose(t, loss.detach(), atol=1e-6)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_gpt_ring_cp2_loss_agrees():
    _run(_gpt_ring_worker, 2)


def _zigzag_worker(rank, world, port):
    """Zigzag sharding must also match full attention exactly (fwd+bwd),
    while levelling the causal blocks each rank computes."""
    _init(rank, world, port)
    from paddlefleetx_amd.parallel.ring import ring_attention, zigzag_slice
    torch.manual_seed(1)
    B, H, S, D = 2, 4, 32, 8
    scale = 1.0 / math.sqrt(D)
    q = torch.randn(B, H, S, D)
    k = torch.randn(B, H, S, D)
    v = torch.randn(B, H, S, D)
    do = torch.randn(B, H, S, D)
    qr = q.clone().requires_grad_(True)
    kr = k.clone().requires_grad_(True)
    vr = v.clone().requires_grad_(True)
    ref = _full_ref(qr, kr, vr, scale)
    ref.backward(do)

    def zz(t):
        return zigzag_slice(t, world, rank, dim=2)

    ql = zz(q).clone().requires_grad_(True)
    kl = zz(k).clone().requires_grad_(True)
    vl = zz(v).clone().requires_grad_(True)
    o = ring_attention(ql, kl, vl, scale, zigzag=True)
    assert torch.allclose(o, zz(ref.detach()), atol=1e-5), \
        (o - zz(ref.detach())).abs().max()
    o.backward(zz(do))
    for name, got, want in (("dq", ql.grad, zz(qr.grad)),
                            ("dk", kl.grad, zz(kr.grad)),
                            ("dv", vl.grad, zz(vr.grad))):
        assert torch.allclose(got, want, atol=1e-4), \
            (name, (got - want).abs().max())
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ring_attention_zigzag_cp2():
    _run(_zigzag_worker, 2)


@pytest.mark.timeout(300)
def test_ring_attention_zigzag_cp4():
    _run(_zigzag_worker, 4)


def _gpt_zigzag_worker(rank, world, port):
    _init(rank, world, port)
    from paddlefleetx_amd.models import build_module
    cfg = {
        "Global": {"global_batch_size": 2},
        "Engine": {"mix_precision": {"enable": False}},
        "Model": {"name": "GPTModule", "vocab_size": 128, "hidden_size": 32,
                  "num_layers": 2, "num_attention_heads": 4,
                  "max_position_embeddings": 32,
                  "hidden_dropout_prob": 0.0,
                  "attention_probs_dropout_prob": 0.0, "fused_attn": True,
                  "cp_backend": "ring", "cp_zigzag": True},
        "Distributed": {"cp_degree": world},
    }
    torch.manual_seed(5)
    mod = build_module(cfg)
    torch.manual_seed(9)
    tokens = torch.randint(0, 128, (2, 32))
    pos = torch.arange(32).unsqueeze(0).repeat(2, 1)
    labels = torch.randint(0, 128, (2, 32))
    mask = torch.ones(2, 32)
    batch = mod.pretreating_batch((tokens, pos, labels, mask))
    loss = mod.training_step(batch)
    loss.backward()
    t = loss.detach().clone()
    dist.broadcast(t, src=0)
    assert torch.allclose(t, loss.detach(), atol=1e-6)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_gpt_ring_zigzag_cp2_loss_agrees():
    _run(_gpt_zigzag_worker, 2)
