"""MoE gates: naive top-k, GShard top-2, Switch top-1.

Reference: ppfleetx/models/language_model/moe/gate/{naive_gate.py:43,
gshard_gate.py:29-72, switch_gate.py:29-74} (fastmoe-derived) and the
DeepSpeed-style top1gating/top2gating in moe_exp/sharded_moe.py:134/226.

All gating bookkeeping (histogram, capacity pruning, random routing) runs
as plain tensor ops — the hot dispatch path (sort + all-to-all) lives in
moe_layer.py.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


class BaseGate(nn.Module):
    def __init__(self, d_model: int, num_experts: int, top_k: int):
        super().__init__()
        self.d_model = d_model
        self.num_experts = num_experts
        self.top_k = top_k
        self.loss: Optional[torch.Tensor] = None

    def get_loss(self) -> Optional[torch.Tensor]:
        return self.loss


class NaiveGate(BaseGate):
    """Linear gate -> top-k, softmax over selected scores; no aux loss
    (naive_gate.py:43)."""

    def __init__(self, d_model: int, num_experts: int, top_k: int = 2):
        super().__init__(d_model, num_experts, top_k)
        self.gate = nn.Linear(d_model, num_experts, bias=True)

    def forward(self, x: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        logits = self.gate(x.float())
        val, idx = torch.topk(logits, k=self.top_k, dim=-1)
        score = F.softmax(val, dim=-1)
        self.loss = None
        return idx, score


def _load_balance_loss(gate_probs: torch.Tensor, assign_idx: torch.Tensor,
                       num_experts: int) -> torch.Tensor:
    """GShard aux loss: E * sum_e( frac_tokens_e * mean_prob_e )
    (sharded_moe.py top2gating; Switch Transformer eq. 4)."""
    T = gate_probs.shape[0]
    me = gate_probs.mean(dim=0)  # mean router prob per expert
    ce = torch.bincount(assign_idx.reshape(-1), minlength=num_experts
                        ).float() / max(1, assign_idx.numel())
    return num_experts * torch.sum(me * ce)


class GShardGate(BaseGate):
    """Top-2 gate with aux load-balance loss, capacity and random second-expert
    routing (gshard_gate.py:29-72)."""

    def __init__(self, d_model: int, num_experts: int, top_k: int = 2,
                 capacity: Tuple[float, float] = (1.2, 2.4),
                 random_routing: bool = True):
        assert top_k == 2, "GShard gate is top-2"
        super().__init__(d_model, num_experts, 2)
        self.gate = nn.Linear(d_model, num_experts, bias=False)
        self.capacity = capacity
        self.random_routing = random_routing

    def forward(self, x: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        logits = self.gate(x.float())
        probs = F.softmax(logits, dim=-1)
        topv, topi = torch.topk(probs, k=2, dim=-1)
        self.loss = _load_balance_loss(probs, topi[:, 0], self.num_experts)
        if self.random_routing and self.training:
            # keep the 2nd expert with prob 2*p2 (gshard paper §3.2)
            rand = torch.rand_like(topv[:, 1])
            drop = rand > (2.0 * topv[:, 1])
            topi = topi.clone()
            topi[:, 1] = torch.where(drop, topi[:, 0], topi[:, 1])
        score = topv / topv.sum(dim=-1, keepdim=True).clamp(min=1e-9)
        return topi, score

    def capacity_for(self, num_tokens: int, training: bool) -> int:
        f = self.capacity[0] if training else self.capacity[1]
        return max(1, int(math.ceil(f * num_tokens / self.num_experts)))


class SwitchGate(BaseGate):
    """Top-1 gate with multiplicative jitter noise + aux loss
    (switch_gate.py:29-74)."""

    def __init__(self, d_model: int, num_experts: int, top_k: int = 1,
                 switch_eps: float = 0.1,
                 capacity: Tuple[float, float] = (1.2, 2.4)):
        assert top_k == 1, "Switch gate is top-1"
        super().__init__(d_model, num_experts, 1)
        self.gate = nn.Linear(d_model, num_experts, bias=False)
        self.switch_eps = switch_eps
        self.capacity = capacity

    def forward(self, x: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        logits = self.gate(x.float())
        if self.training and self.switch_eps > 0:
            noise = torch.empty_like(logits).uniform_(
                1.0 - self.switch_eps, 1.0 + self.switch_eps)
            logits = logits * noise
        probs = F.softmax(logits, dim=-1)
        topv, topi = torch.topk(probs, k=1, dim=-1)
        self.loss = _load_balance_loss(probs, topi[:, 0], self.num_experts)
        return topi, torch.ones_like(topv)

    def capacity_for(self, num_tokens: int, training: bool) -> int:
        f = self.capacity[0] if training else self.capacity[1]
        return max(1, int(math.ceil(f * num_tokens / self.num_experts)))


def build_gate(name: str, d_model: int, num_experts: int, top_k: int,
               **kw) -> BaseGate:
    table = {"naive": NaiveGate, "gshard": GShardGate, "switch": SwitchGate}
    if name not in table:
        raise ValueError(f"unknown gate {name!r} (have {sorted(table)})")
    return table[name](d_model, num_experts, top_k=top_k, **kw)
