"""Context parallelism (Ulysses): sequence sharding via head<->seq all-to-all.

Generalizes the reference's DAP axis-swap primitive
(ppfleetx/distributed/protein_folding/dap.py:244-379 `_all_to_all`,
`row_to_col`/`col_to_row`) into a first-class long-context axis: each CP
rank holds [B, S/N, H] activations; inside attention one all-to-all
redistributes to [B, S, h/N] per rank (full sequence, 1/N of the heads),
the gfx950 flash kernel runs unchanged, and the inverse all-to-all
restores the sequence sharding. On the 8xMI355X xGMI mesh the all-to-all
is all-pairs traffic — exactly what the fully-connected 7-link topology
serves best — and 288 GB HBM/GPU makes seq 32k+ feasible at 6.7B.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from paddlefleetx_amd.parallel.env import get_hcg

__all__ = ["all_to_all_4d", "seq_to_head", "head_to_seq",
           "UlyssesAttention", "scatter_to_cp", "gather_from_cp",
           "cp_allreduce_sum"]


def _cp_group():
    return get_hcg().get_context_parallel_group()


def _a2a_4d(x: torch.Tensor, scatter_dim: int, gather_dim: int,
            group) -> torch.Tensor:
    """All-to-all that scatters `scatter_dim` and gathers `gather_dim`."""
    world = dist.get_world_size(group)
    if world == 1:
        return x
    # split along scatter_dim into world chunks, exchange, concat on gather
    inp = [c.contiguous() for c in torch.chunk(x, world, dim=scatter_dim)]
    out = [torch.empty_like(c) for c in inp]
    if dist.get_backend(group) == "gloo":
        # pairwise isend/irecv fallback for CPU tests
        my = dist.get_rank(group)
        ranks = dist.get_process_group_ranks(group)
        reqs = []
        for peer in range(world):
            if peer == my:
                out[peer] = inp[peer]
                continue
            reqs.append(dist.irecv(out[peer], src=ranks[peer], group=group))
            reqs.append(dist.isend(inp[peer], dst=ranks[peer], group=group))
        for r in reqs:
            r.wait()
    else:
        dist.all_to_all(out, inp, group=group)
    return torch.cat(out, dim=gather_dim)


class _AllToAll4D(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, scatter_dim, gather_dim, group):
        ctx.scatter_dim, ctx.gather_dim, ctx.group = \
            scatter_dim, gather_dim, group
        return _a2a_4d(x, scatter_dim, gather_dim, group)

    @staticmethod
    def backward(ctx, gy):
        gx = _a2a_4d(gy.contiguous(), ctx.gather_dim, ctx.scatter_dim,
                     ctx.group)
        return gx, None, None, None


def all_to_all_4d(x, scatter_dim: int, gather_dim: int, group=None):
    g = group if group is not None else _cp_group().group
    if g is None:
        return x
    return _AllToAll4D.apply(x, scatter_dim, gather_dim, g)


def seq_to_head(qkv_like: torch.Tensor, group=None) -> torch.Tensor:
    """[B, S/N, H, D] -> [B, S, H/N, D] (dap.py row_to_col analog)."""
    return all_to_all_4d(qkv_like, scatter_dim=2, gather_dim=1, group=group)


def head_to_seq(o: torch.Tensor, group=None) -> torch.Tensor:
    """[B, S, H/N, D] -> [B, S/N, H, D] (dap.py col_to_row analog)."""
    return all_to_all_4d(o, scatter_dim=1, gather_dim=2, group=group)


class UlyssesAttention(torch.nn.Module):
    """Wraps an attention kernel call with the Ulysses A2A pair.

    Input/output: [B, S_local, H, D] sequence-sharded over the cp group.
    """

    def __init__(self, scale: Optional[float] = None, causal: bool = True):
        super().__init__()
        self.scale = scale
        self.causal = causal

    def forward(self, q, k, v):
        gi = _cp_group()
        g = gi.group
        world = gi.world_size
        if world > 1:
            q = seq_to_head(q, g)
            k = seq_to_head(k, g)
            v = seq_to_head(v, g)
        # [B, S, h_local, D] -> [B, h, S, D] for the kernel
        qt, kt, vt = (t.transpose(1, 2) for t in (q, k, v))
        if qt.is_cuda and qt.dtype == torch.bfloat16 and \
                qt.shape[-1] in (64, 128):
            from paddlefleetx_amd.ops import flash_attention
            o = flash_attention(qt, kt, vt, causal=self.causal,
                                scale=self.scale)
        else:
            import math
            scale = self.scale or 1.0 / math.sqrt(qt.shape[-1])
            s = torch.matmul(qt.float(), kt.float().transpose(-1, -2)) * scale
            if self.causal:
                S1, S2 = s.shape[-2], s.shape[-1]
                mask = torch.ones(S1, S2, dtype=torch.bool,
                                  device=s.device).tril(S2 - S1)
                s = s.masked_fill(~mask, float("-inf"))
            p = torch.softmax(s, dim=-1)
            o = torch.matmul(p, vt.float()).to(qt.dtype)
        o = o.transpose(1, 2)  # [B, S, h_local, D]
        if world > 1:
            o = head_to_seq(o, g)
        return o


class _ScatterToCP(torch.autograd.Function):
    """Slice the local sequence chunk fwd; all-gather grads bwd."""

    @staticmethod
    def forward(ctx, x, dim):
        gi = _cp_group()
        ctx.dim, ctx.world, ctx.rank = dim, gi.world_size, gi.rank
        ctx.group = gi.group
        if gi.world_size == 1:
            return x
        return torch.chunk(x, gi.world_size, dim=dim)[gi.rank].contiguous()

    @staticmethod
    def backward(ctx, gy):
        if ctx.world == 1:
            return gy, None
        parts = [torch.empty_like(gy) for _ in range(ctx.world)]
        dist.all_gather(parts, gy.contiguous(), group=ctx.group)
        return torch.cat(parts, dim=ctx.dim), None


class _GatherFromCP(torch.autograd.Function):
    """All-gather the sequence fwd; slice grads bwd."""

    @staticmethod
    def forward(ctx, x, dim):
        gi = _cp_group()
        ctx.dim, ctx.world, ctx.rank = dim, gi.world_size, gi.rank
        if gi.world_size == 1:
            return x
        parts = [torch.empty_like(x) for _ in range(gi.world_size)]
        dist.all_gather(parts, x.contiguous(), group=gi.group)
        return torch.cat(parts, dim=dim)

    @staticmethod
    def backward(ctx, gy):
        if ctx.world == 1:
            return gy, None
        return torch.chunk(gy, ctx.world, dim=ctx.dim)[ctx.rank].contiguous(), \
            None


def scatter_to_cp(x, dim: int = 1):
    return _ScatterToCP.apply(x, dim)


def gather_from_cp(x, dim: int = 1):
    return _GatherFromCP.apply(x, dim)


class _CPAllReduceSum(torch.autograd.Function):
    """Differentiable sum-allreduce over the cp group (loss reduction):
    forward sums across ranks, backward is identity (each rank's
    contribution gets the same upstream gradient)."""

    @staticmethod
    def forward(ctx, x):
        gi = _cp_group()
        if gi.world_size > 1 and dist.is_initialized():
            x = x.clone()
            dist.all_reduce(x, group=gi.group)
        return x

    @staticmethod
    def backward(ctx, gy):
        return gy


def cp_allreduce_sum(x):
    return _CPAllReduceSum.apply(x)
