"""Linear layers with weight-gradient accumulation fused into the GEMM.

With FusedAdamW, every param's `.grad` is a pre-attached view into a flat
bucket (optims/optimizer.py). Plain autograd still materializes each
micro-batch's dW in a fresh tensor and then launches an add into that
view (AccumulateGrad) — ~1.5k full-tensor adds per step on GPT-6.7B
(profiled 3.5% of step time). Here the backward computes

    weight.grad.addmm_(dy^T, x)        # beta=1 — accumulate IN the GEMM
    bias.grad += colsum(dy)            # one fused column-sum kernel

and returns None for the weight. AccumulateGrad still executes with the
None gradient — it launches no kernel but DOES fire the post-accumulate
hooks, so the DP-overlap bucket countdown (optims/optimizer.py
enable_overlap) keeps working unchanged. The reference gets the same
effect from Paddle's fused_linear param storages
(tensor_fusion_helper.py:90-106); this is the MI355X expression of it.
"""

from __future__ import annotations

from typing import Callable, Optional

import torch
import torch.nn.functional as F

__all__ = ["fused_linear", "fused_bias_add", "set_wgrad_fusion",
           "wgrad_fusion_enabled"]

_STATE = {"enabled": False, "notify": None}


def set_wgrad_fusion(enabled: bool,
                     notify: Optional[Callable[[torch.nn.Parameter], None]]
                     = None) -> None:
    """Engine hook: enable the fused-wgrad path (FusedAdamW bucket mode
    with grads in param dtype). `notify(param)` is called when a fused
    param's gradient for the current backward is complete."""
    _STATE["enabled"] = bool(enabled)
    _STATE["notify"] = notify


def wgrad_fusion_enabled() -> bool:
    return _STATE["enabled"]


def _grad_view(p: torch.nn.Parameter) -> Optional[torch.Tensor]:
    g = p.grad
    if g is None or g.dtype != p.dtype or g.shape != p.shape:
        return None
    return g


class _FusedWgradLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        ctx.weight_ref = weight
        ctx.bias_ref = bias
        return F.linear(x, weight, bias)

    @staticmethod
    @torch.autograd.function.once_differentiable
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        dyc = dy.contiguous()
        dy2 = dyc.reshape(-1, dyc.shape[-1])
        x2 = x.reshape(-1, x.shape[-1])
        dx = dyc.matmul(w).view(x.shape)
        # NOTE: AccumulateGrad still executes for the weight even though
        # we return None (verified: fires the post-accumulate hooks used
        # by the DP overlap countdown, launches NO add kernel) — so the
        # bucket bookkeeping needs no separate notification.
        wref = ctx.weight_ref
        g = _grad_view(wref)
        if g is not None:
            g.addmm_(dy2.t(), x2)  # dW += dy^T x, accumulated in-GEMM
        else:  # bucket view lost (shouldn't happen) — fall back loudly
            with torch.no_grad():
                wref.grad = dy2.t().mm(x2) if wref.grad is None \
                    else wref.grad + dy2.t().mm(x2)
        bref = ctx.bias_ref
        if bref is not None:
            bg = _grad_view(bref)
            db = _colsum(dy2)
            if bg is not None:
                bg.add_(db.to(bg.dtype))
            else:
                with torch.no_grad():
                    bref.grad = db.to(bref.dtype) if bref.grad is None \
                        else bref.grad + db.to(bref.dtype)
        return dx, None, None


def _colsum(x2: torch.Tensor) -> torch.Tensor:
    if x2.is_cuda:
        from paddlefleetx_amd.ops import hip_ext
        return hip_ext().colsum(x2)
    return x2.float().sum(0)


def _fusable(weight: torch.nn.Parameter, bias) -> bool:
    if not _STATE["enabled"] or not torch.is_grad_enabled():
        return False
    # measured exception (gpurun bench_fused_ops): the fused-QKV wgrad
    # shape [3h, h] loses hipBLASLt's SplitK at beta=1 (addmm 777us vs
    # mm 687us + 33us add) — keep plain autograd for that one shape
    if weight.dim() == 2 and weight.shape[0] == 3 * weight.shape[1]:
        return False
    if not isinstance(weight, torch.nn.Parameter) or not weight.requires_grad:
        return False
    if _grad_view(weight) is None:
        return False
    if bias is not None and (not isinstance(bias, torch.nn.Parameter)
                             or _grad_view(bias) is None):
        return False
    return True


def fused_linear(x, weight, bias=None):
    """F.linear with in-GEMM weight-grad accumulation when the engine has
    armed the fused path; plain F.linear otherwise."""
    if _fusable(weight, bias):
        return _FusedWgradLinear.apply(x, weight, bias)
    return F.linear(x, weight, bias)


class _FusedBiasAdd(torch.autograd.Function):
    """y + bias (broadcast over rows) with the bias grad computed by the
    fused column-sum kernel and accumulated into the bucket view."""

    @staticmethod
    def forward(ctx, y, bias):
        ctx.bias_ref = bias
        return y + bias

    @staticmethod
    @torch.autograd.function.once_differentiable
    def backward(ctx, dy):
        bref = ctx.bias_ref
        bg = _grad_view(bref)
        dy2 = dy.contiguous().reshape(-1, dy.shape[-1])
        db = _colsum(dy2)
        if bg is not None:
            bg.add_(db.to(bg.dtype))
        else:
            with torch.no_grad():
                bref.grad = db.to(bref.dtype) if bref.grad is None \
                    else bref.grad + db.to(bref.dtype)
        return dy, None


def fused_bias_add(y, bias):
    if bias is None:
        return y
    if _fusable(bias, None):
        return _FusedBiasAdd.apply(y, bias)
    return y + bias
