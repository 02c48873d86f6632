"""LR schedules (reference ppfleetx/optims/lr_scheduler.py:22-120).

Schedulers here are framework-free step counters returning the lr value;
the engine passes the value into FusedAdamW.step(lr=...). Supports the
token-increment stepping mode (`lr.step(epoch=global_batch_size)`,
eager_engine.py:354-357) via `step(increment)`.
"""

from __future__ import annotations

import math

__all__ = ["CosineAnnealingWithWarmupDecay", "LinearDecayWithWarmup",
           "ConstantLR", "build_lr_scheduler"]


class _Scheduler:
    def __init__(self):
        self.num = 0

    def step(self, increment: int = 1):
        self.num += increment

    def get_lr(self) -> float:
        raise NotImplementedError

    def state_dict(self):
        return {"num": self.num}

    def load_state_dict(self, sd):
        self.num = sd["num"]


class CosineAnnealingWithWarmupDecay(_Scheduler):
    def __init__(self, max_lr: float, min_lr: float, warmup_rate: float = 0.01,
                 decay_steps: int = 360000, warmup_steps: int = None, **unused):
        super().__init__()
        self.max_lr, self.min_lr = max_lr, min_lr
        self.decay_steps = decay_steps
        self.warmup_steps = warmup_steps if warmup_steps is not None \
            else int(warmup_rate * decay_steps)

    def get_lr(self) -> float:
        if self.warmup_steps > 0 and self.num <= self.warmup_steps:
            return self.max_lr * self.num / self.warmup_steps
        if self.num > self.decay_steps:
            return self.min_lr
        ratio = (self.num - self.warmup_steps) / max(
            1, self.decay_steps - self.warmup_steps)
        coeff = 0.5 * (1.0 + math.cos(math.pi * ratio))
        return self.min_lr + coeff * (self.max_lr - self.min_lr)


class LinearDecayWithWarmup(_Scheduler):
    def __init__(self, learning_rate: float, total_steps: int,
                 warmup: float = 0.1, **unused):
        super().__init__()
        self.lr = learning_rate
        self.total = total_steps
        self.warmup_steps = int(warmup * total_steps) if warmup < 1 else int(warmup)

    def get_lr(self) -> float:
        if self.num < self.warmup_steps:
            return self.lr * self.num / max(1, self.warmup_steps)
        return max(0.0, self.lr * (self.total - self.num)
                   / max(1, self.total - self.warmup_steps))


class ConstantLR(_Scheduler):
    def __init__(self, learning_rate: float = 1e-4, **unused):
        super().__init__()
        self.lr = learning_rate

    def get_lr(self) -> float:
        return self.lr


class ViTLRScheduler(_Scheduler):
    """Linear warmup + cosine/linear decay (reference lr_scheduler.py
    ViTLRScheduler; decay_type 'cosine' | 'linear')."""

    def __init__(self, learning_rate: float, warmup_steps: int = 10000,
                 total_steps: int = 300000, decay_type: str = "cosine",
                 **unused):
        super().__init__()
        self.lr = learning_rate
        self.warmup_steps = warmup_steps
        self.total_steps = total_steps
        self.decay_type = decay_type

    def get_lr(self) -> float:
        if self.warmup_steps > 0 and self.num < self.warmup_steps:
            return self.lr * self.num / max(1, self.warmup_steps)
        ratio = (self.num - self.warmup_steps) / max(
            1, self.total_steps - self.warmup_steps)
        ratio = min(1.0, ratio)
        if self.decay_type == "cosine":
            return self.lr * 0.5 * (1.0 + math.cos(math.pi * ratio))
        return self.lr * (1.0 - ratio)


class CosineAnnealingDecay(_Scheduler):
    """Plain cosine decay (reference MoCo configs: CosineAnnealingDecay)."""

    def __init__(self, learning_rate: float, T_max: int = 100,
                 eta_min: float = 0.0, **unused):
        super().__init__()
        self.lr = learning_rate
        self.T_max = max(1, int(T_max))
        self.eta_min = eta_min

    def get_lr(self) -> float:
        ratio = min(1.0, self.num / self.T_max)
        return self.eta_min + 0.5 * (self.lr - self.eta_min) * (
            1.0 + math.cos(math.pi * ratio))


class MultiStepDecay(_Scheduler):
    def __init__(self, learning_rate: float, milestones=(30, 60, 90),
                 gamma: float = 0.1, **unused):
        super().__init__()
        self.lr = learning_rate
        self.milestones = sorted(milestones)
        self.gamma = gamma

    def get_lr(self) -> float:
        n = sum(1 for m in self.milestones if self.num >= m)
        return self.lr * (self.gamma ** n)


def build_lr_scheduler(cfg) -> _Scheduler:
    cfg = dict(cfg or {})
    name = cfg.pop("name", "ConstantLR")
    table = {
        "CosineAnnealingWithWarmupDecay": CosineAnnealingWithWarmupDecay,
        "LinearDecayWithWarmup": LinearDecayWithWarmup,
        "ConstantLR": ConstantLR,
        "ViTLRScheduler": ViTLRScheduler,
        "MultiStepDecay": MultiStepDecay,
        "CosineAnnealingDecay": CosineAnnealingDecay,
    }
    if name not in table:
        raise ValueError(f"unknown lr scheduler {name}")
    return table[name](**cfg)
