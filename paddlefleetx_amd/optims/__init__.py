"""Optimizer/LR builders (reference ppfleetx/optims/__init__.py:29-74)."""

from __future__ import annotations

from paddlefleetx_amd.optims.lr_scheduler import (ConstantLR,
                                                  CosineAnnealingWithWarmupDecay,
                                                  LinearDecayWithWarmup,
                                                  build_lr_scheduler)
from paddlefleetx_amd.optims.optimizer import AdamW, FusedAdamW

__all__ = ["build_optimizer", "build_lr_scheduler", "FusedAdamW", "AdamW",
           "CosineAnnealingWithWarmupDecay", "LinearDecayWithWarmup", "ConstantLR"]

def _momentum(named_params, lr=0.1, weight_decay=0.0, momentum=0.9,
              **unused):
    """SGD+momentum for the vision configs (reference optims Momentum)."""
    import torch
    params = [p for _, p in named_params]
    return torch.optim.SGD(params, lr=lr, momentum=momentum,
                           weight_decay=weight_decay)


_OPTIMIZERS = {"FusedAdamW": FusedAdamW, "AdamW": AdamW,
               "Momentum": _momentum}


def build_optimizer(cfg, model, lr_value: float = None, **extra):
    cfg = dict(cfg or {})
    name = cfg.pop("name", "FusedAdamW")
    cfg.pop("lr", None)
    grad_clip = cfg.pop("grad_clip", None)
    if isinstance(grad_clip, dict):
        grad_clip = grad_clip.get("clip_norm", 1.0)
    if name not in _OPTIMIZERS:
        raise ValueError(f"unknown optimizer {name}")
    kwargs = {k: v for k, v in cfg.items()
              if k in ("weight_decay", "beta1", "beta2", "epsilon",
                       "multi_precision", "tensor_fusion", "grad_dtype")}
    if name == "FusedAdamW":
        return _OPTIMIZERS[name](
            model.named_parameters(), lr=lr_value or 1e-4,
            grad_clip=grad_clip,
            sharding_group=extra.get("sharding_group"),
            sharding_stage=extra.get("sharding_stage", 1), **kwargs)
    return _OPTIMIZERS[name](model.named_parameters(), lr=lr_value or 1e-4,
                             **kwargs)
