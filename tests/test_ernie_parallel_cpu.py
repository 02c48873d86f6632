"""ERNIE TP2 and PP2 variants must match the single-process model
(gloo, CPU). Reference parity: ErnieModelHybrid
(ernie/dygraph/hybrid_model.py:167) and ErnieForPretrainingPipe (:796).
"""

import os
import tempfile

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

REPO = os.path.join(os.path.dirname(__file__), "..")

CFG = dict(vocab_size=128, hidden_size=32, num_hidden_layers=2,
           num_attention_heads=4, intermediate_size=64,
           hidden_dropout_prob=0.0, attention_probs_dropout_prob=0.0,
           max_position_embeddings=32, type_vocab_size=2)


def _init(rank, world, port, mp_deg=1, pp=1):
    import sys
    sys.path.insert(0, REPO)
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world),
                      LOCAL_RANK=str(rank))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from paddlefleetx_amd.parallel.env import set_hcg, set_seed
    from paddlefleetx_amd.parallel.topology import HybridTopology
    hcg = HybridTopology(mp=mp_deg, pp=pp)
    set_hcg(hcg)
    set_seed(1234)
    return hcg


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


def _batch():
    torch.manual_seed(11)
    B, S = 4, 16
    input_ids = torch.randint(2, 128, (B, S))
    token_type_ids = torch.zeros(B, S, dtype=torch.long)
    labels = torch.full((B, S), -1, dtype=torch.long)
    labels[:, 2:6] = torch.randint(2, 128, (B, 4))
    nsp = torch.randint(0, 2, (B,))
    return input_ids, token_type_ids, labels, nsp


# ---------------------------------------------------------------------------
# TP2
# ---------------------------------------------------------------------------

def _shard_tp(tp_model, sd, mpd, r):
    """Load a single-card ErnieForPretraining state dict into the TP model,
    slicing by the documented [q_shard; k_shard; v_shard] qkv layout."""
    H = CFG["hidden_size"]
    Hm = H // mpd
    I = CFG["intermediate_size"]
    V = CFG["vocab_size"]
    out = {}
    for name, p in tp_model.state_dict().items():
        w = sd[name]
        if name.endswith("word_embeddings.weight") or \
                name.endswith("decoder_weight"):  # tied vocab shard
            w = w[r * V // mpd:(r + 1) * V // mpd]
        elif ".attn.qkv." in name:
            w = torch.cat([w[c * H + r * Hm:(c * H + (r + 1) * Hm)]
                           for c in range(3)], dim=0)
        elif ".attn.out_proj.weight" in name:
            w = w[:, r * Hm:(r + 1) * Hm]
        elif ".fc1." in name or name.endswith("fc1_bias"):
            w = w[r * I // mpd:(r + 1) * I // mpd]
        elif ".fc2.weight" in name:
            w = w[:, r * I // mpd:(r + 1) * I // mpd]
        elif name.endswith("decoder_bias"):
            w = w[r * V // mpd:(r + 1) * V // mpd]
        assert w.shape == p.shape, (name, w.shape, p.shape)
        out[name] = w
    tp_model.load_state_dict(out)


def _ref_state(tmpdir):
    """Single-card model + reference loss, saved for the workers."""
    import sys
    sys.path.insert(0, REPO)
    from paddlefleetx_amd.models.ernie.model import (
        ErnieForPretraining, ErnieModel, ErniePretrainingCriterion)
    from paddlefleetx_amd.parallel.env import set_hcg
    from paddlefleetx_amd.parallel.topology import HybridTopology
    set_hcg(HybridTopology())
    torch.manual_seed(3)
    m = ErnieForPretraining(ErnieModel(**CFG))
    input_ids, tt, labels, nsp = _batch()
    pred, rel = m(input_ids, tt)
    mlm, nsp_l = ErniePretrainingCriterion()(pred, rel, labels, nsp)
    loss = mlm + nsp_l
    loss.backward()
    torch.save({"sd": m.state_dict(), "loss": loss.detach(),
                "pooler_grad": m.ernie.pooler.dense.weight.grad.clone()},
               os.path.join(tmpdir, "ref.pt"))


def _tp_worker(rank, world, port, tmpdir):
    _init(rank, world, port, mp_deg=2)
    from paddlefleetx_amd.models.ernie.model import (
        ErnieForPretraining, ErnieModel, ErniePretrainingCriterion)
    ref = torch.load(os.path.join(tmpdir, "ref.pt"), weights_only=False)
    m = ErnieForPretraining(ErnieModel(**CFG))
    _shard_tp(m, ref["sd"], 2, rank)
    input_ids, tt, labels, nsp = _batch()
    pred, rel = m(input_ids, tt)
    assert pred.shape[-1] == CFG["vocab_size"] // 2  # vocab-parallel logits
    mlm, nsp_l = ErniePretrainingCriterion()(pred, rel, labels, nsp)
    loss = mlm + nsp_l
    assert torch.allclose(loss, ref["loss"], atol=2e-5), (loss, ref["loss"])
    loss.backward()
    g = m.ernie.pooler.dense.weight.grad
    assert torch.allclose(g, ref["pooler_grad"], atol=1e-4), \
        (g - ref["pooler_grad"]).abs().max()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_ernie_tp2_matches_single():
    with tempfile.TemporaryDirectory() as tmpdir:
        _ref_state(tmpdir)
        _run(_tp_worker, 2, (tmpdir,))


# ---------------------------------------------------------------------------
# PP2 (pipe descs + tied embedding), against the pp=1 pipe model
# ---------------------------------------------------------------------------

def _pipe_ref_worker(rank, world, port, tmpdir):
    _init(rank, world, port, pp=1)
    from paddlefleetx_amd.models.ernie.pipeline_model import (
        ErnieForPretrainingPipe, ErniePipeCriterion)
    torch.manual_seed(7)
    m = ErnieForPretrainingPipe(**CFG)
    for i, layer in enumerate(m.layers):
        torch.save(layer.state_dict(),
                   os.path.join(tmpdir, f"layer_{m._layer_desc_idx[i]}.pt"))
    loss = m.forward_backward_pipeline(_batch(), ErniePipeCriterion(),
                                       accumulate_steps=2)
    emb_g = m.layers[0].word_embeddings.weight.grad.clone()
    torch.save({"loss": loss, "emb_grad": emb_g},
               os.path.join(tmpdir, "pref.pt"))
    dist.destroy_process_group()


def _pipe2_worker(rank, world, port, tmpdir):
    _init(rank, world, port, pp=2)
    from paddlefleetx_amd.models.ernie.pipeline_model import (
        ErnieForPretrainingPipe, ErniePipeCriterion)
    m = ErnieForPretrainingPipe(**CFG)
    for i, layer in enumerate(m.layers):
        sd = torch.load(
            os.path.join(tmpdir, f"layer_{m._layer_desc_idx[i]}.pt"),
            weights_only=False)
        layer.load_state_dict(sd)
    loss = m.forward_backward_pipeline(_batch(), ErniePipeCriterion(),
                                       accumulate_steps=2)
    ref = torch.load(os.path.join(tmpdir, "pref.pt"), weights_only=False)
    assert torch.allclose(loss, ref["loss"], atol=1e-5), (loss, ref["loss"])
    if rank == 0:
        g = m.layers[0].word_embeddings.weight.grad
        assert torch.allclose(g, ref["emb_grad"], atol=1e-4), \
            (g - ref["emb_grad"]).abs().max()
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_ernie_pipe2_matches_single():
    with tempfile.TemporaryDirectory() as tmpdir:
        _run(_pipe_ref_worker, 1, (tmpdir,))
        _run(_pipe2_worker, 2, (tmpdir,))


def _pipe_vpp_worker(rank, world, port, tmpdir):
    """ERNIE pipeline with interleaved virtual stages (pp2 x V2 over 4
    encoder layers) matches the pp1 reference."""
    _init(rank, world, port, pp=2)
    from paddlefleetx_amd.models.ernie.pipeline_model import (
        ErnieForPretrainingPipe, ErniePipeCriterion)
    cfg4 = dict(CFG, num_hidden_layers=4)
    m = ErnieForPretrainingPipe(virtual_pp_degree=2, **cfg4)
    assert m.num_virtual == 2
    for i, layer in enumerate(m.layers):
        sd = torch.load(
            os.path.join(tmpdir, f"vl_{m._layer_desc_idx[i]}.pt"),
            weights_only=False)
        layer.load_state_dict(sd)
    loss = m.forward_backward_pipeline(_batch(), ErniePipeCriterion(),
                                       accumulate_steps=2)
    ref = torch.load(os.path.join(tmpdir, "vref.pt"), weights_only=False)
    assert torch.allclose(loss, ref["loss"], atol=1e-5), (loss, ref["loss"])
    dist.barrier()
    dist.destroy_process_group()


def _pipe_vpp_ref_worker(rank, world, port, tmpdir):
    _init(rank, world, port, pp=1)
    from paddlefleetx_amd.models.ernie.pipeline_model import (
        ErnieForPretrainingPipe, ErniePipeCriterion)
    torch.manual_seed(7)
    cfg4 = dict(CFG, num_hidden_layers=4)
    m = ErnieForPretrainingPipe(**cfg4)
    for i, layer in enumerate(m.layers):
        torch.save(layer.state_dict(),
                   os.path.join(tmpdir, f"vl_{m._layer_desc_idx[i]}.pt"))
    loss = m.forward_backward_pipeline(_batch(), ErniePipeCriterion(),
                                       accumulate_steps=2)
    torch.save({"loss": loss}, os.path.join(tmpdir, "vref.pt"))
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_ernie_pipe_vpp2_matches_single():
    with tempfile.TemporaryDirectory() as tmpdir:
        _run(_pipe_vpp_ref_worker, 1, (tmpdir,))
        _run(_pipe_vpp_worker, 2, (tmpdir,))
