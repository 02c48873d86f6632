"""MoE-aware global-norm gradient clipping.

Reference: ppfleetx/optims/grad_clip.py:27-170 ClipGradForMOEByGlobalNorm —
expert-parameter grad norms are reduced over the expert-parallel group
(each EP rank holds different experts, so their norm contributions must be
summed across the group), then combined with the shared-parameter norm.
"""

from __future__ import annotations

from typing import Iterable, Optional

import torch
import torch.distributed as dist


def _sq_norm(params: Iterable[torch.nn.Parameter]) -> torch.Tensor:
    total = None
    for p in params:
        g = getattr(p, "main_grad", None)
        if g is None:
            g = p.grad
        if g is None:
            continue
        n = g.float().pow(2).sum()
        total = n if total is None else total + n
    if total is None:
        return torch.zeros(())
    return total


@torch.no_grad()
def clip_grad_for_moe_by_global_norm(parameters, clip_norm: float,
                                     moe_group=None) -> float:
    """Clip grads in place; returns the pre-clip global norm.

    Expert params (p.is_expert) have their squared-norm allreduce-summed over
    `moe_group` before combining with the shared-param norm (grad_clip.py:
    27-170).
    """
    params = [p for p in parameters if p.requires_grad]
    expert = [p for p in params if getattr(p, "is_expert", False)]
    shared = [p for p in params if not getattr(p, "is_expert", False)]

    shared_sq = _sq_norm(shared)
    expert_sq = _sq_norm(expert)
    if moe_group is not None and dist.is_initialized() and \
            dist.get_world_size(moe_group) > 1:
        t = expert_sq.clone().detach()
        dist.all_reduce(t, group=moe_group)
        expert_sq = t
    global_norm = torch.sqrt(shared_sq + expert_sq)
    scale = clip_norm / (float(global_norm) + 1e-6)
    if scale < 1.0:
        for p in params:
            g = getattr(p, "main_grad", None)
            if g is None:
                g = p.grad
            if g is not None:
                g.mul_(scale)
    return float(global_norm)
