"""DAP (Dynamic Axial Parallelism) + BP (Branch Parallelism) for folding.

Reference: ppfleetx/distributed/protein_folding/dap.py (scatter/gather
:60-240, `_all_to_all` :244, `row_to_col` :358 [N,S,R,C -> N,R,S,C
axis-swap], `col_to_row` :379) and bp.py :39-95 (broadcast fwd +
grad-allreduce). The all-to-all primitive is shared with the Ulysses CP
axis (parallel/cp.py) — DAP is its [msa, res] instantiation.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from paddlefleetx_amd.parallel.cp import _AllToAll4D
from paddlefleetx_amd.parallel.env import get_hcg

__all__ = ["scatter", "gather", "row_to_col", "col_to_row",
           "bp_broadcast", "bp_grad_allreduce"]


def _dap_group():
    # DAP runs over the model-parallel group (reference scg.py dap group)
    return get_hcg().get_model_parallel_group()


class _Scatter(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, dim, group_info):
        ctx.dim = dim
        ctx.gi = group_info
        if group_info.world_size == 1:
            return x
        return torch.chunk(x, group_info.world_size,
                           dim=dim)[group_info.rank].contiguous()

    @staticmethod
    def backward(ctx, gy):
        gi = ctx.gi
        if gi.world_size == 1:
            return gy, None, None
        parts = [torch.empty_like(gy) for _ in range(gi.world_size)]
        dist.all_gather(parts, gy.contiguous(), group=gi.group)
        return torch.cat(parts, dim=ctx.dim), None, None


class _Gather(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, dim, group_info):
        ctx.dim = dim
        ctx.gi = group_info
        if group_info.world_size == 1:
            return x
        parts = [torch.empty_like(x) for _ in range(group_info.world_size)]
        dist.all_gather(parts, x.contiguous(), group=group_info.group)
        return torch.cat(parts, dim=dim)

    @staticmethod
    def backward(ctx, gy):
        gi = ctx.gi
        if gi.world_size == 1:
            return gy, None, None
        return torch.chunk(gy, gi.world_size,
                           dim=ctx.dim)[gi.rank].contiguous(), None, None


def scatter(x, dim: int, group=None):
    return _Scatter.apply(x, dim, group or _dap_group())


def gather(x, dim: int, group=None):
    return _Gather.apply(x, dim, group or _dap_group())


def row_to_col(x, group=None):
    """[N, S/k, R, C] -> [N, S, R/k, C] (dap.py:358)."""
    gi = group or _dap_group()
    if gi.world_size == 1:
        return x
    return _AllToAll4D.apply(x, 2, 1, gi.group)


def col_to_row(x, group=None):
    """[N, S, R/k, C] -> [N, S/k, R, C] (dap.py:379)."""
    gi = group or _dap_group()
    if gi.world_size == 1:
        return x
    return _AllToAll4D.apply(x, 1, 2, gi.group)


# ---------------------------------------------------------------------------
# BP: 2-way branch parallel (bp.py:39-95)
# ---------------------------------------------------------------------------

class _BpBroadcast(torch.autograd.Function):
    """Broadcast fwd from src; allreduce grads bwd so every branch's
    contribution reaches the shared parameters."""

    @staticmethod
    def forward(ctx, x, src, group_info):
        ctx.gi = group_info
        if group_info.world_size > 1 and dist.is_initialized():
            x = x.contiguous()
            dist.broadcast(x, src=group_info.ranks[src],
                           group=group_info.group)
        return x

    @staticmethod
    def backward(ctx, gy):
        gi = ctx.gi
        if gi.world_size > 1 and dist.is_initialized():
            gy = gy.contiguous()
            dist.all_reduce(gy, group=gi.group)
        return gy, None, None


def bp_broadcast(x, src: int = 0, group=None):
    return _BpBroadcast.apply(x, src, group or _dap_group())


@torch.no_grad()
def bp_grad_allreduce(params, group=None):
    """Manual branch-grad sync (bp.py grad allreduce over param list)."""
    gi = group or _dap_group()
    if gi.world_size == 1 or not dist.is_initialized():
        return
    for p in params:
        if p.grad is not None:
            dist.all_reduce(p.grad, group=gi.group)
            p.grad.div_(gi.world_size)
