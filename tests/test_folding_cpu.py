"""Protein folding: Evoformer components + DAP/BP comm primitives."""

import pytest
import torch
import torch.distributed as dist

from paddlefleetx_amd.parallel.env import set_hcg
from paddlefleetx_amd.parallel.topology import HybridTopology
from tests.test_distributed_cpu import _init, _run


@pytest.fixture(autouse=True)
def _env():
    set_hcg(HybridTopology())
    yield


def test_gated_attention_shapes_and_bias():
    from paddlefleetx_amd.models.protein_folding import GatedAttention
    torch.manual_seed(0)
    attn = GatedAttention(16, 16, 2, 8, 16, gating=True)
    x = torch.randn(2, 5, 16)
    y = attn(x)
    assert y.shape == (2, 5, 16)
    # an attention bias changes the output
    bias = torch.randn(1, 2, 5, 5) * 5
    y_b = attn(x, bias=bias)
    assert not torch.allclose(y, y_b, atol=1e-5)
    # gate at init is sigmoid(1): disabling gating changes the scale
    attn.gating = False
    y_ng = attn(x)
    assert not torch.allclose(y, y_ng, atol=1e-5)


def test_evoformer_iteration_shapes_and_grads():
    from paddlefleetx_amd.models.protein_folding import EvoformerIteration
    torch.manual_seed(1)
    blk = EvoformerIteration(msa_dim=16, pair_dim=16, num_heads=2,
                             head_dim=8)
    msa = torch.randn(1, 4, 6, 16, requires_grad=True)   # [B, S, R, C]
    pair = torch.randn(1, 6, 6, 16, requires_grad=True)  # [B, R, R, C]
    msa2, pair2 = blk(msa, pair)
    assert msa2.shape == msa.shape and pair2.shape == pair.shape
    (msa2.sum() + pair2.sum()).backward()
    assert msa.grad is not None and pair.grad is not None


def test_triangle_multiplication_outgoing_vs_incoming():
    from paddlefleetx_amd.models.protein_folding import \
        TriangleMultiplication
    torch.manual_seed(2)
    tm_out = TriangleMultiplication(8, 8, outgoing=True)
    tm_in = TriangleMultiplication(8, 8, outgoing=False)
    tm_in.load_state_dict(tm_out.state_dict())
    z = torch.randn(1, 5, 5, 8)
    a, b = tm_out(z), tm_in(z)
    assert a.shape == z.shape
    assert not torch.allclose(a, b)  # different edge contractions


def test_outer_product_mean_pair_shape():
    from paddlefleetx_amd.models.protein_folding import OuterProductMean
    op = OuterProductMean(16, 24, hidden=8)
    msa = torch.randn(2, 3, 7, 16)
    pair = op(msa)
    assert pair.shape == (2, 7, 7, 24)


def _dap_worker(rank, world, port):
    hcg = _init(rank, world, port, mp_deg=2)
    from paddlefleetx_amd.parallel.dap import (col_to_row, gather,
                                               row_to_col, scatter)
    torch.manual_seed(0)
    full = torch.randn(1, 4, 6, 8)  # [N, S, R, C]
    g = hcg.get_model_parallel_group()
    local = scatter(full, dim=1, group=g)
    assert local.shape == (1, 2, 6, 8)
    back = gather(local, dim=1, group=g)
    assert torch.allclose(back, full, atol=1e-6)
    # axis swap: S-sharded -> R-sharded
    col = row_to_col(local, group=g)
    expect = torch.chunk(full, world, dim=2)[rank]
    assert torch.allclose(col, expect, atol=1e-6), (col - expect).abs().max()
    row = col_to_row(col, group=g)
    assert torch.allclose(row, local, atol=1e-6)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dap_axis_swaps():
    _run(_dap_worker, 2)


def _bp_worker(rank, world, port):
    hcg = _init(rank, world, port, mp_deg=2)
    from paddlefleetx_amd.parallel.dap import bp_broadcast
    g = hcg.get_model_parallel_group()
    x = torch.full((4,), float(rank), requires_grad=True)
    y = bp_broadcast(x, src=0, group=g)
    assert torch.all(y == 0.0)  # rank 0's value everywhere
    y.sum().backward()
    # grads allreduced: each rank contributes ones -> sum = world
    assert torch.all(x.grad == world)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_bp_broadcast_grad_allreduce():
    _run(_bp_worker, 2)
