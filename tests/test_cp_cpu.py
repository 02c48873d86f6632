"""Context parallel (Ulysses): a2a primitives + GPT cp2 == single rank."""

import pytest
import torch
import torch.distributed as dist

from tests.test_distributed_cpu import _init, _run


def _cp_init(rank, world, port):
    import os
    import sys
    REPO = __import__("tests.test_distributed_cpu",
                      fromlist=["REPO"]).REPO
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    hcg = HybridTopology(cp=world)
    set_hcg(hcg)
    set_seed(1234)
    return hcg


def _a2a_worker(rank, world, port):
    hcg = _cp_init(rank, world, port)
    from paddlefleetx_amd.parallel.cp import head_to_seq, seq_to_head
    torch.manual_seed(0)  # same full tensor on both ranks
    B, S, H, D = 2, 8, 4, 4
    full = torch.randn(B, S, H, D)
    local = torch.chunk(full, world, dim=1)[rank]  # [B, S/2, H, D]
    g = hcg.get_context_parallel_group().group
    swapped = seq_to_head(local, g)               # [B, S, H/2, D]
    expect = torch.chunk(full, world, dim=2)[rank]
    assert torch.allclose(swapped, expect, atol=1e-6), \
        (swapped - expect).abs().max()
    back = head_to_seq(swapped, g)
    assert torch.allclose(back, local, atol=1e-6)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_cp_a2a_roundtrip():
    _run(_a2a_worker, 2)


def _ulysses_worker(rank, world, port):
    hcg = _cp_init(rank, world, port)
    from paddlefleetx_amd.parallel.cp import UlyssesAttention
    torch.manual_seed(1)
    B, S, H, D = 1, 16, 4, 8
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, H, D)
    v = torch.randn(B, S, H, D)
    ua = UlyssesAttention(causal=True)
    local = ua(torch.chunk(q, world, 1)[rank].contiguous(),
               torch.chunk(k, world, 1)[rank].contiguous(),
               torch.chunk(v, world, 1)[rank].contiguous())
    # single-rank reference: plain causal attention on full tensors
    import math
    scale = 1.0 / math.sqrt(D)
    qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v))
    s = torch.matmul(qt, kt.transpose(-1, -2)) * scale
    mask = torch.ones(S, S, dtype=torch.bool).tril()
    s = s.masked_fill(~mask, float("-inf"))
    ref = torch.matmul(torch.softmax(s, -1), vt).transpose(1, 2)
    expect = torch.chunk(ref, world, dim=1)[rank]
    assert torch.allclose(local, expect, atol=1e-5), \
        (local - expect).abs().max()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ulysses_attention_matches_full():
    _run(_ulysses_worker, 2)


def _gpt_cp_worker(rank, world, port):
    hcg = _cp_init(rank, world, port)
    from paddlefleetx_amd.models import build_module
    cfg = {
        "Global": {"global_batch_size": 2},
        "Engine": {"mix_precision": {"enable": False}},
        "Model": {"name": "GPTModule", "vocab_size": 128, "hidden_size": 32,
                  "num_layers": 2, "num_attention_heads": 4,
                  "max_position_embeddings": 32,
                  "hidden_dropout_prob": 0.0,
                  "attention_probs_dropout_prob": 0.0, "fused_attn": False},
        "Distributed": {"cp_degree": world},
    }
    torch.manual_seed(5)  # identical init on both ranks
    mod = build_module(cfg)
    torch.manual_seed(9)
    tokens = torch.randint(0, 128, (2, 32))
    pos = torch.arange(32).unsqueeze(0).repeat(2, 1)
    labels = torch.randint(0, 128, (2, 32))
    mask = torch.ones(2, 32)
    batch = mod.pretreating_batch((tokens, pos, labels, mask))
    assert batch[0].shape == (2, 16)  # sliced to the local chunk
    loss = mod.training_step(batch)
    loss.backward()

    # single-rank twin computes the same loss
    from paddlefleetx_amd.parallel.env import set_hcg
    from paddlefleetx_amd.parallel.topology import HybridTopology

    # compare against a rank-0 broadcast of the loss: both ranks must agree
    t = loss.detach().clone()
    dist.broadcast(t, src=0)
    assert torch.allclose(t, loss.detach(), atol=1e-6)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_gpt_cp2_global_loss_agrees():
    _run(_gpt_cp_worker, 2)
