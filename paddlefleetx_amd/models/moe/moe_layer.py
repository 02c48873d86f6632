"""MoE layer: gate -> sort-dispatch -> all-to-all -> expert FFN -> combine.

Reference: ppfleetx/models/language_model/moe/moe_layer.py:33-235
(MoEScatter/MoEGather comm_ops.py:28-118, count/alltoall bookkeeping
moe/utils.py:26-51, capacity limit :110-126, experts_fwd :195-208, weighted
combine :226-228) and ExpertLayer (single_model.py:56-80).

MI355X-native design: the expert-parallel all-to-all maps directly onto the
fully-connected 7-link xGMI mesh (every GPU has a dedicated link to every
peer, EP8 = all-pairs traffic), issued as one RCCL all_to_all_single with
uneven splits. Token permutation uses torch.argsort (rocPRIM radix sort on
GPU) instead of the reference's custom _assign_pos kernel; the permute
itself is index_select (gather kernel) and the combine is a weighted
index_add — all differentiable, so only the all-to-all needs a custom
autograd Function (gradient = the reverse all-to-all).
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from paddlefleetx_amd.models.moe.gate import BaseGate, build_gate
from paddlefleetx_amd.ops import bias_gelu
from paddlefleetx_amd.parallel.env import get_hcg


def _a2a_exchange(x: torch.Tensor, out_splits: List[int],
                  in_splits: List[int], group) -> torch.Tensor:
    """Uneven all-to-all. RCCL path: one all_to_all_single; gloo fallback
    (CPU tests): isend/irecv pairs."""
    world = dist.get_world_size(group)
    out = x.new_empty((sum(out_splits),) + x.shape[1:])
    if dist.get_backend(group) != "gloo":
        dist.all_to_all_single(out, x.contiguous(),
                               output_split_sizes=out_splits,
                               input_split_sizes=in_splits, group=group)
        return out
    # gloo: pairwise exchange
    my = dist.get_rank(group)
    in_off = [0]
    for s in in_splits:
        in_off.append(in_off[-1] + s)
    out_off = [0]
    for s in out_splits:
        out_off.append(out_off[-1] + s)
    reqs = []
    xc = x.contiguous()
    global_ranks = dist.get_process_group_ranks(group)
    for peer in range(world):
        if peer == my:
            out[out_off[peer]:out_off[peer + 1]] = \
                xc[in_off[peer]:in_off[peer + 1]]
            continue
        if out_splits[peer] > 0:
            reqs.append(dist.irecv(out[out_off[peer]:out_off[peer + 1]],
                                   src=global_ranks[peer], group=group))
        if in_splits[peer] > 0:
            reqs.append(dist.isend(
                xc[in_off[peer]:in_off[peer + 1]].contiguous(),
                dst=global_ranks[peer], group=group))
    for r in reqs:
        r.wait()
    return out


class _DispatchGather(torch.autograd.Function):
    """Token dispatch: sort slots by expert and gather their token rows.

    GPU: ONE fused HIP pass (histogram + atomic rank + row scatter,
    csrc/moe.hip) replacing argsort + index_select (reference
    global_scatter / _assign_pos, moe/comm_ops.py:28-71). CPU: the
    stable-argsort equivalent. Backward is the index_add of the row
    gather (the permutation itself carries no gradient)."""

    @staticmethod
    def forward(ctx, xf, flat_expert, num_experts, top_k):
        if xf.is_cuda:
            from paddlefleetx_amd.ops import hip_ext
            dispatched, sel_sorted, counts = hip_ext().moe_dispatch(
                xf, flat_expert, num_experts, top_k)
        else:
            sel = (flat_expert >= 0).nonzero(as_tuple=True)[0]
            active = flat_expert[sel]
            perm = torch.argsort(active, stable=True)
            sel_sorted = sel[perm]
            counts = torch.bincount(active[perm], minlength=num_experts)
            dispatched = xf.index_select(0, sel_sorted // top_k)
        ctx.save_for_backward(sel_sorted // top_k)
        ctx.T = xf.shape[0]
        return dispatched, sel_sorted, counts

    @staticmethod
    def backward(ctx, d_disp, _dsel, _dcnt):
        (token_of_slot,) = ctx.saved_tensors
        dxf = d_disp.new_zeros(ctx.T, d_disp.shape[-1])
        dxf.index_add_(0, token_of_slot, d_disp)
        return dxf, None, None, None


def _dispatch_gather(xf, flat_expert, num_experts, top_k):
    return _DispatchGather.apply(xf, flat_expert, num_experts, top_k)


class _AllToAll(torch.autograd.Function):
    """Differentiable uneven all-to-all (comm_ops.py:28-118 global_scatter/
    global_gather collapse into this one primitive + local permutes)."""

    @staticmethod
    def forward(ctx, x, out_splits, in_splits, group):
        ctx.out_splits, ctx.in_splits, ctx.group = out_splits, in_splits, group
        return _a2a_exchange(x, out_splits, in_splits, group)

    @staticmethod
    def backward(ctx, gy):
        gx = _a2a_exchange(gy.contiguous(), ctx.in_splits, ctx.out_splits,
                           ctx.group)
        return gx, None, None, None


def all_to_all(x, out_splits, in_splits, group):
    if group is None or dist.get_world_size(group) == 1:
        return x
    return _AllToAll.apply(x, out_splits, in_splits, group)


class ExpertLayer(nn.Module):
    """One expert FFN: up -> fused bias-gelu -> down (single_model.py:56-80)."""

    def __init__(self, d_model: int, d_hidden: int,
                 dtype: Optional[torch.dtype] = None):
        super().__init__()
        self.up = nn.Linear(d_model, d_hidden, bias=False, dtype=dtype)
        self.up_bias = nn.Parameter(torch.zeros(d_hidden, dtype=dtype))
        self.down = nn.Linear(d_hidden, d_model, bias=True, dtype=dtype)
        for p in self.parameters():
            p.is_expert = True  # MoE grad clip separates these (grad_clip.py:27)

    def forward(self, x):
        return self.down(bias_gelu(self.up(x), self.up_bias))


class MoELayer(nn.Module):
    """Expert-parallel MoE FFN (moe_layer.py:33-235).

    Each EP rank holds `num_experts / ep_world` local experts. Forward:
      1. gate -> (topk_idx [T,k], score [T,k]) + aux loss
      2. optional capacity drop (moe/utils.py:110-126 limit_by_capacity)
      3. stable sort of the T*k slots by global expert id; counts histogram
      4. exchange counts over EP group, all-to-all the selected tokens
      5. run local experts segment-wise, all-to-all back
      6. combine: weighted index_add into [T, d]
    """

    def __init__(self, d_model: int, d_hidden: int, num_experts: int,
                 gate: str = "gshard", top_k: int = 2,
                 capacity_factor: Optional[float] = None,
                 dtype: Optional[torch.dtype] = None,
                 gate_kwargs: Optional[dict] = None):
        super().__init__()
        hcg = get_hcg()
        self.ep_group_info = hcg.get_expert_parallel_group()
        self.ep_world = self.ep_group_info.world_size
        assert num_experts % self.ep_world == 0, (
            f"num_experts {num_experts} not divisible by ep {self.ep_world}")
        self.num_experts = num_experts
        self.num_local_experts = num_experts // self.ep_world
        self.top_k = top_k
        self.capacity_factor = capacity_factor
        if isinstance(gate, BaseGate):
            self.gate = gate
        else:
            self.gate = build_gate(gate, d_model, num_experts, top_k,
                                   **(gate_kwargs or {}))
        self.experts = nn.ModuleList([
            ExpertLayer(d_model, d_hidden, dtype=dtype)
            for _ in range(self.num_local_experts)])
        for p in self.gate.parameters():
            p.is_gate = True

    # -- capacity ----------------------------------------------------------
    def _apply_capacity(self, topk_idx, score):
        """Drop slots whose per-expert arrival position exceeds capacity;
        dropped slots keep their token (identity) via zero score."""
        if self.capacity_factor is None:
            return topk_idx, score, None
        T = topk_idx.shape[0]
        cap = max(1, int(self.capacity_factor * T * self.top_k /
                         self.num_experts))
        flat = topk_idx.reshape(-1)
        # position of each slot within its expert's queue (stable)
        order = torch.argsort(flat, stable=True)
        ranks = torch.empty_like(order)
        seg = torch.bincount(flat, minlength=self.num_experts)
        pos_in_seg = torch.cat([torch.arange(int(c), device=flat.device)
                                for c in seg]) if seg.sum() else order
        ranks[order] = pos_in_seg
        keep = (ranks < cap).reshape(T, self.top_k)
        score = score * keep.to(score.dtype)
        # renormalize remaining weights (gshard keeps raw weights; we follow)
        return topk_idx, score, keep

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        orig_shape = x.shape
        d = orig_shape[-1]
        xf = x.reshape(-1, d)
        T = xf.shape[0]

        topk_idx, score = self.gate(xf)  # [T,k] int64, [T,k] fp32
        topk_idx, score, keep = self._apply_capacity(topk_idx, score)

        flat_expert = topk_idx.reshape(-1)  # [T*k]
        if keep is not None:
            # route dropped slots to a sentinel so they are excluded
            flat_expert = torch.where(keep.reshape(-1), flat_expert,
                                      torch.full_like(flat_expert, -1))
        dispatched, sel_sorted, counts = _dispatch_gather(
            xf, flat_expert, self.num_experts, self.top_k)
        token_of_slot = sel_sorted // self.top_k
        # tokens leaving to each EP peer (num_local_experts each)
        send_per_rank = counts.reshape(self.ep_world, -1).sum(dim=1)

        if self.ep_world > 1:
            recv_counts = torch.empty_like(counts)
            dist.all_to_all_single(
                recv_counts, counts, group=self.ep_group_info.group) \
                if dist.get_backend(self.ep_group_info.group) != "gloo" else \
                self._gloo_count_exchange(recv_counts, counts)
            # recv_counts[r*L + e] = #tokens from peer r for my local expert e?
            # layout: counts is [world * local] indexed by global expert id;
            # after a2a each peer's slice for MY experts arrives.
            my = self.ep_group_info.rank
            L = self.num_local_experts
            recv_matrix = recv_counts.reshape(self.ep_world, L)
            recv_per_rank = recv_matrix.sum(dim=1)
            in_splits = [int(c) for c in send_per_rank]
            out_splits = [int(c) for c in recv_per_rank]
        else:
            recv_matrix = counts.reshape(1, -1)
            in_splits = out_splits = [int(counts.sum())]

        dispatched = all_to_all(dispatched, out_splits, in_splits,
                                self.ep_group_info.group
                                if self.ep_world > 1 else None)

        # segment-wise expert execution: arrivals are grouped by source rank,
        # each group ordered by local expert id -> regroup per local expert
        L = self.num_local_experts
        if self.ep_world > 1:
            # build order: for each local expert, concat each source rank's seg
            seg_sizes = recv_matrix  # [world, L]
            # offsets of (rank, expert) segment in arrival buffer
            arrival_off = torch.zeros(self.ep_world, L, dtype=torch.long)
            flatoff = 0
            for r in range(self.ep_world):
                for e in range(L):
                    arrival_off[r, e] = flatoff
                    flatoff += int(seg_sizes[r, e])
            pieces = []
            bounds = [0]
            for e in range(L):
                for r in range(self.ep_world):
                    n = int(seg_sizes[r, e])
                    if n:
                        o = int(arrival_off[r, e])
                        pieces.append(torch.arange(o, o + n))
                bounds.append(bounds[-1] + int(seg_sizes[:, e].sum()))
            if pieces:
                gather_idx = torch.cat(pieces).to(dispatched.device)
                grouped = dispatched.index_select(0, gather_idx)
            else:
                gather_idx = torch.empty(0, dtype=torch.long,
                                         device=dispatched.device)
                grouped = dispatched[:0]
        else:
            grouped = dispatched
            bounds = [0]
            for e in range(L):
                bounds.append(bounds[-1] + int(recv_matrix[0, e]))

        outs = []
        for e in range(L):
            seg = grouped[bounds[e]:bounds[e + 1]]
            outs.append(self.experts[e](seg) if seg.shape[0] else seg)
        expert_out = torch.cat(outs, dim=0) if outs else grouped

        if self.ep_world > 1:
            # un-group back to arrival order, then reverse all-to-all
            ungrouped = expert_out.new_empty(expert_out.shape)
            if gather_idx.numel():
                ungrouped.index_copy_(0, gather_idx, expert_out)
            returned = all_to_all(ungrouped, in_splits, out_splits,
                                  self.ep_group_info.group)
        else:
            returned = expert_out

        # combine: weighted sum over the k slots of each token
        slot_w = score.reshape(-1)[sel_sorted].to(returned.dtype)
        combined = torch.zeros_like(xf)
        combined.index_add_(0, token_of_slot, returned * slot_w.unsqueeze(1))
        return combined.reshape(orig_shape)

    def _gloo_count_exchange(self, recv, counts):
        world = self.ep_world
        L = self.num_local_experts
        gathered = [torch.empty_like(counts) for _ in range(world)]
        dist.all_gather(gathered, counts, group=self.ep_group_info.group)
        my = self.ep_group_info.rank
        for r in range(world):
            recv[r * L:(r + 1) * L] = gathered[r][my * L:(my + 1) * L]
        return recv
