"""Second MoE implementation surface: einsum-dispatch gating (DeepSpeed
style).

Reference: ppfleetx/models/language_model/moe_exp/sharded_moe.py —
top1gating :134, top2gating :226, TopKGate :300, MOELayer :379 with the
one-hot einsum dispatch/combine (:87-117) and the _AllToAll PyLayer
(:66-84). The primary implementation (moe_layer.py) uses the fused HIP
index dispatch; this one keeps the einsum formulation for API parity and
as the dense-dispatch reference the fused path is tested against.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from paddlefleetx_amd.models.moe.moe_layer import ExpertLayer, all_to_all


def _one_hot(idx: torch.Tensor, num_classes: int) -> torch.Tensor:
    return F.one_hot(idx, num_classes=num_classes).to(torch.float32)


def top1gating(logits: torch.Tensor, capacity_factor: float = 1.0,
               min_capacity: int = 4, noisy_gate_policy: Optional[str] = None
               ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor,
                          torch.Tensor]:
    """Returns (aux_loss, combine_weights [T, E, C], dispatch_mask
    [T, E, C] bool, exp_counts [E]). Reference sharded_moe.py:134-223."""
    T, E = logits.shape
    if noisy_gate_policy == "RSample" and logits.requires_grad:
        logits_for_route = logits + torch.randn_like(logits) / E
    else:
        logits_for_route = logits
    gates = logits.softmax(dim=-1)
    idx = logits_for_route.argmax(dim=-1)
    mask1 = _one_hot(idx, E)                         # [T, E]
    exp_counts = mask1.sum(dim=0)
    capacity = max(min_capacity, int(capacity_factor * T / E))

    # aux (load-balance) loss: E * <fraction routed> . <mean gate prob>
    me = gates.mean(dim=0)
    ce = mask1.mean(dim=0)
    aux = (me * ce).sum() * E

    # position within the expert queue; drop overflow
    locations = torch.cumsum(mask1, dim=0) - 1.0
    mask1 = mask1 * (locations < capacity).float()
    loc1 = (locations * mask1).sum(dim=-1).long()    # [T]

    gates1 = (gates * mask1).sum(dim=-1)             # [T] routed prob
    combine = gates1[:, None, None] * mask1[:, :, None] * \
        _one_hot(loc1, capacity)[:, None, :]         # [T, E, C]
    dispatch = combine.bool()
    return aux, combine, dispatch, exp_counts


def top2gating(logits: torch.Tensor, capacity_factor: float = 1.0,
               min_capacity: int = 4
               ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor,
                          torch.Tensor]:
    """Top-2 variant (reference sharded_moe.py:226-297)."""
    T, E = logits.shape
    gates = logits.softmax(dim=-1)
    idx1 = gates.argmax(dim=-1)
    mask1 = _one_hot(idx1, E)
    # second expert from the masked logits
    logits_wo1 = logits.masked_fill(mask1.bool(), float("-inf"))
    idx2 = logits_wo1.argmax(dim=-1)
    mask2 = _one_hot(idx2, E)
    capacity = max(min_capacity, int(2 * capacity_factor * T / E))

    me = gates.mean(dim=0)
    ce = mask1.mean(dim=0)
    aux = (me * ce).sum() * E

    loc1 = torch.cumsum(mask1, dim=0) - 1.0
    loc2 = torch.cumsum(mask2, dim=0) - 1.0 + mask1.sum(dim=0, keepdim=True)
    mask1 = mask1 * (loc1 < capacity).float()
    mask2 = mask2 * (loc2 < capacity).float()
    l1 = (loc1 * mask1).sum(dim=-1).long()
    l2 = (loc2 * mask2).sum(dim=-1).long()

    g1 = (gates * mask1).sum(dim=-1)
    g2 = (gates * mask2).sum(dim=-1)
    denom = (g1 + g2).clamp_min(torch.finfo(gates.dtype).eps)
    g1, g2 = g1 / denom, g2 / denom
    combine = (g1[:, None, None] * mask1[:, :, None] *
               _one_hot(l1, capacity)[:, None, :] +
               g2[:, None, None] * mask2[:, :, None] *
               _one_hot(l2, capacity)[:, None, :])
    return aux, combine, combine.bool(), (mask1 + mask2).sum(dim=0)


class TopKGate(nn.Module):
    """Gate wrapper (reference sharded_moe.py:300-376)."""

    def __init__(self, d_model: int, num_experts: int, k: int = 1,
                 capacity_factor: float = 1.0, min_capacity: int = 4,
                 noisy_gate_policy: Optional[str] = None):
        super().__init__()
        assert k in (1, 2)
        self.wg = nn.Linear(d_model, num_experts, bias=False)
        self.k = k
        self.capacity_factor = capacity_factor
        self.min_capacity = min_capacity
        self.noisy_gate_policy = noisy_gate_policy

    def forward(self, x):
        logits = self.wg(x.float())
        if self.k == 1:
            return top1gating(logits, self.capacity_factor,
                              self.min_capacity, self.noisy_gate_policy)
        return top2gating(logits, self.capacity_factor, self.min_capacity)


class ShardedMoELayer(nn.Module):
    """Einsum-dispatch MoE layer (reference MOELayer :379: dispatch =
    einsum("tec,tm->ecm"), a2a over EP, expert FFNs, reverse a2a,
    combine = einsum("tec,ecm->tm"))."""

    def __init__(self, d_model: int, d_hidden: int, num_experts: int,
                 k: int = 1, capacity_factor: float = 1.0,
                 ep_group=None, dtype: Optional[torch.dtype] = None,
                 **gate_kwargs):
        super().__init__()
        self.gate = TopKGate(d_model, num_experts, k, capacity_factor,
                             **gate_kwargs)
        self.ep_group = ep_group
        self.ep_world = (ep_group.world_size
                         if ep_group is not None and
                         hasattr(ep_group, "world_size") else 1)
        assert num_experts % max(1, self.ep_world) == 0
        self.num_experts = num_experts
        self.num_local = num_experts // max(1, self.ep_world)
        self.experts = nn.ModuleList([
            ExpertLayer(d_model, d_hidden, dtype=dtype)
            for _ in range(self.num_local)])
        self.last_aux_loss: Optional[torch.Tensor] = None

    def forward(self, x):
        orig = x.shape
        d = orig[-1]
        xf = x.reshape(-1, d)
        aux, combine, dispatch, _ = self.gate(xf)
        self.last_aux_loss = aux
        combine = combine.to(x.dtype)
        # [E, C, M] dense dispatch
        dispatched = torch.einsum("tec,tm->ecm", dispatch.to(x.dtype), xf)
        if self.ep_world > 1:
            E, C, M = dispatched.shape
            g = self.ep_group.group if hasattr(self.ep_group, "group") \
                else self.ep_group
            flat = dispatched.reshape(-1, M)
            splits = [E // self.ep_world * C] * self.ep_world
            flat = all_to_all(flat, splits, splits, g)
            dispatched = flat.reshape(E, C, M)  # peers' slices stacked
        # expert compute on the local slice(s)
        E, C, M = dispatched.shape
        chunks = dispatched.reshape(self.ep_world * self.num_local, C, M) \
            if self.ep_world > 1 else dispatched
        outs = []
        for i in range(chunks.shape[0]):
            outs.append(self.experts[i % self.num_local](chunks[i]))
        expert_out = torch.stack(outs, dim=0)
        if self.ep_world > 1:
            g = self.ep_group.group if hasattr(self.ep_group, "group") \
                else self.ep_group
            flat = expert_out.reshape(-1, M)
            splits = [E // self.ep_world * C] * self.ep_world
            flat = all_to_all(flat, splits, splits, g)
            expert_out = flat.reshape(E, C, M)
        out = torch.einsum("tec,ecm->tm", combine, expert_out)
        return out.reshape(orig)
