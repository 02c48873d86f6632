"""Megatron-style sequence parallelism over the TP group.

Activations between layers are [s/mp, b, h]; column linears all-gather the
sequence dim before the GEMM, row linears reduce-scatter after it — same
comm volume as TP's allreduce but LayerNorm/dropout run on sharded
activations. Reference: ppfleetx gpt/dygraph/sequence_parallel_utils.py
(ScatterOp/GatherOp/AllGatherOp/ReduceScatterOp :84-137,
ColumnSequenceParallelLinear :222-304, RowSequenceParallelLinear :307-398,
LN-param grad allreduce hooks :147-212).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from paddlefleetx_amd.ops.linear import fused_bias_add, fused_linear

from paddlefleetx_amd.parallel.env import get_hcg
from paddlefleetx_amd.parallel.rng import model_parallel_rng

__all__ = [
    "ScatterOp", "GatherOp", "AllGatherOp", "ReduceScatterOp",
    "scatter_to_sp_region", "gather_from_sp_region",
    "ColumnSequenceParallelLinear", "RowSequenceParallelLinear",
    "mark_as_sp_param", "allreduce_sp_param_grads",
]


def _mp():
    return get_hcg().get_model_parallel_group()


def _split_dim0(x, g):
    chunk = x.shape[0] // g.world_size
    return x[g.rank * chunk:(g.rank + 1) * chunk].contiguous()


def _all_gather_dim0(x, g):
    x = x.contiguous()
    out = torch.empty((g.world_size * x.shape[0],) + tuple(x.shape[1:]),
                      dtype=x.dtype, device=x.device)
    dist.all_gather_into_tensor(out, x, group=g.group)
    return out


def _reduce_scatter_dim0(x, g):
    x = x.contiguous()
    out = torch.empty((x.shape[0] // g.world_size,) + tuple(x.shape[1:]),
                      dtype=x.dtype, device=x.device)
    dist.reduce_scatter_tensor(out, x, group=g.group)
    return out


class ScatterOp(torch.autograd.Function):
    """fwd: take own seq chunk; bwd: all-gather grads."""

    @staticmethod
    def forward(ctx, x):
        g = _mp()
        if g.world_size == 1:
            return x
        return _split_dim0(x, g)

    @staticmethod
    def backward(ctx, gy):
        g = _mp()
        if g.world_size == 1:
            return gy
        return _all_gather_dim0(gy, g)


class GatherOp(torch.autograd.Function):
    """fwd: all-gather seq dim; bwd: take own chunk."""

    @staticmethod
    def forward(ctx, x):
        g = _mp()
        if g.world_size == 1:
            return x
        return _all_gather_dim0(x, g)

    @staticmethod
    def backward(ctx, gy):
        g = _mp()
        if g.world_size == 1:
            return gy
        return _split_dim0(gy, g)


class AllGatherOp(torch.autograd.Function):
    """fwd: all-gather; bwd: reduce-scatter (input of SP column linear)."""

    @staticmethod
    def forward(ctx, x):
        g = _mp()
        if g.world_size == 1:
            return x
        return _all_gather_dim0(x, g)

    @staticmethod
    def backward(ctx, gy):
        g = _mp()
        if g.world_size == 1:
            return gy
        return _reduce_scatter_dim0(gy, g)


class ReduceScatterOp(torch.autograd.Function):
    """fwd: reduce-scatter; bwd: all-gather (output of SP row linear)."""

    @staticmethod
    def forward(ctx, x):
        g = _mp()
        if g.world_size == 1:
            return x
        return _reduce_scatter_dim0(x, g)

    @staticmethod
    def backward(ctx, gy):
        g = _mp()
        if g.world_size == 1:
            return gy
        return _all_gather_dim0(gy, g)


def scatter_to_sp_region(x: torch.Tensor) -> torch.Tensor:
    """[b, s, h] embedding output -> [s/mp, b, h] (hybrid_model.py:727-735)."""
    x = x.transpose(0, 1).contiguous()  # [s, b, h]
    return ScatterOp.apply(x)


def gather_from_sp_region(x: torch.Tensor) -> torch.Tensor:
    """[s/mp, b, h] -> [b, s, h] full sequence (final GatherOp :891-892)."""
    x = GatherOp.apply(x)
    return x.transpose(0, 1).contiguous()


class ColumnSequenceParallelLinear(nn.Module):
    """allgather(x over seq) -> x @ W^T (+b). W sharded on out dim."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True,
                 dtype: Optional[torch.dtype] = None, init_std: float = 0.02):
        super().__init__()
        g = _mp()
        assert out_features % g.world_size == 0
        self.out_per_rank = out_features // g.world_size
        self.weight = nn.Parameter(
            torch.empty(self.out_per_rank, in_features, dtype=dtype))
        self.weight.is_mp = True
        self.weight.partition_dim = 0
        with model_parallel_rng():
            nn.init.normal_(self.weight, mean=0.0, std=init_std)
        if bias:
            self.bias = nn.Parameter(torch.zeros(self.out_per_rank, dtype=dtype))
            self.bias.is_mp = True
            self.bias.partition_dim = 0
        else:
            self.bias = None

    def forward(self, x):
        # x: [s/mp, b, in] -> [s, b, out/mp]
        x = AllGatherOp.apply(x)
        return fused_linear(x, self.weight, self.bias)


class RowSequenceParallelLinear(nn.Module):
    """x_local @ W^T -> reduce_scatter over seq. W sharded on in dim."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True,
                 dtype: Optional[torch.dtype] = None, init_std: float = 0.02):
        super().__init__()
        g = _mp()
        assert in_features % g.world_size == 0
        self.in_per_rank = in_features // g.world_size
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_per_rank, dtype=dtype))
        self.weight.is_mp = True
        self.weight.partition_dim = 1
        with model_parallel_rng():
            nn.init.normal_(self.weight, mean=0.0, std=init_std)
        if bias:
            self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype))
            mark_as_sp_param(self.bias)  # applied post-scatter: grad needs mp allreduce
        else:
            self.bias = None

    def forward(self, x):
        # x: [s, b, in/mp] -> [s/mp, b, out]
        y = fused_linear(x, self.weight, None)
        y = ReduceScatterOp.apply(y)
        if self.bias is not None:
            y = fused_bias_add(y, self.bias)
        return y


def mark_as_sp_param(p: torch.nn.Parameter) -> None:
    """Params whose grads are computed on seq shards (LN weights, post-RS
    biases) need an allreduce over the mp group before the optimizer step
    (reference sequence_parallel_utils.py:147-212)."""
    p.sequence_parallel = True


def allreduce_sp_param_grads(model: nn.Module) -> None:
    g = _mp()
    if g.world_size == 1:
        return
    grads = [p.grad for p in model.parameters()
             if getattr(p, "sequence_parallel", False) and p.grad is not None]
    if not grads:
        return
    flat = torch._utils._flatten_dense_tensors(grads)
    dist.all_reduce(flat, group=g.group)
    for buf, synced in zip(grads, torch._utils._unflatten_dense_tensors(flat, grads)):
        buf.copy_(synced)
