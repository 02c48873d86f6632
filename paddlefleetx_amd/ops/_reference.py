"""Plain-PyTorch fp32 reference implementations of every HIP op.

These are the numerics twins the GPU kernels are tested against
(SURVEY.md §4 test strategy), and the CPU execution path for tests on
non-GPU hosts. They are NOT used on a GPU box — ops dispatch to the HIP
extension there and fail loudly if it is missing.
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F


def layernorm_fwd(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
                  eps: float) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (y, mean, rstd); stats in fp32, y in x.dtype. x: [N, H]."""
    xf = x.float()
    mean = xf.mean(dim=-1)
    var = xf.var(dim=-1, unbiased=False)
    rstd = torch.rsqrt(var + eps)
    y = (xf - mean[:, None]) * rstd[:, None] * weight.float() + bias.float()
    return y.to(x.dtype), mean, rstd


def layernorm_bwd(dy: torch.Tensor, x: torch.Tensor, weight: torch.Tensor,
                  mean: torch.Tensor, rstd: torch.Tensor
                  ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    xf, dyf, wf = x.float(), dy.float(), weight.float()
    H = x.shape[-1]
    xhat = (xf - mean[:, None]) * rstd[:, None]
    dyw = dyf * wf
    c1 = dyw.mean(dim=-1, keepdim=True)
    c2 = (dyw * xhat).mean(dim=-1, keepdim=True)
    dx = (dyw - c1 - xhat * c2) * rstd[:, None]
    dw = (dyf * xhat).sum(dim=0)
    db = dyf.sum(dim=0)
    return dx.to(x.dtype), dw, db


def layernorm_fwd_residual(x: torch.Tensor, res: torch.Tensor,
                           weight: torch.Tensor, bias: torch.Tensor,
                           eps: float):
    """LN(x + res); returns (y, mean, rstd, sum) with sum = x + res
    in x.dtype (the fused residual stream)."""
    s = (x.float() + res.float()).to(x.dtype)
    y, mean, rstd = layernorm_fwd(s, weight, bias, eps)
    return y, mean, rstd, s


def layernorm_bwd_residual(dy: torch.Tensor, s: torch.Tensor,
                           weight: torch.Tensor, mean: torch.Tensor,
                           rstd: torch.Tensor, dsum: torch.Tensor):
    """LN bwd with the residual-stream gradient fused into dx."""
    dx, dw, db = layernorm_bwd(dy, s, weight, mean, rstd)
    dx = (dx.float() + dsum.float()).to(dx.dtype)
    return dx, dw, db


def rmsnorm_fwd(x: torch.Tensor, weight: torch.Tensor, eps: float
                ) -> Tuple[torch.Tensor, torch.Tensor]:
    xf = x.float()
    rstd = torch.rsqrt(xf.pow(2).mean(dim=-1) + eps)
    y = xf * rstd[:, None] * weight.float()
    return y.to(x.dtype), rstd


def rmsnorm_bwd(dy: torch.Tensor, x: torch.Tensor, weight: torch.Tensor,
                rstd: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    xf, dyf, wf = x.float(), dy.float(), weight.float()
    H = x.shape[-1]
    xhat = xf * rstd[:, None]
    dyw = dyf * wf
    c = (dyw * xhat).mean(dim=-1, keepdim=True)
    dx = (dyw - xhat * c) * rstd[:, None]
    dw = (dyf * xhat).sum(dim=0)
    return dx.to(x.dtype), dw


def bias_gelu_fwd(x: torch.Tensor, bias: Optional[torch.Tensor]) -> torch.Tensor:
    xf = x.float()
    if bias is not None:
        xf = xf + bias.float()
    # tanh approximation (matches the fused kernel)
    y = 0.5 * xf * (1.0 + torch.tanh(0.7978845608028654 * (xf + 0.044715 * xf ** 3)))
    return y.to(x.dtype)


def bias_gelu_bwd(dy: torch.Tensor, x: torch.Tensor, bias: Optional[torch.Tensor]
                  ) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    xf = x.float()
    if bias is not None:
        xf = xf + bias.float()
    k = 0.7978845608028654
    t = torch.tanh(k * (xf + 0.044715 * xf ** 3))
    dgelu = 0.5 * (1.0 + t) + 0.5 * xf * (1.0 - t * t) * k * (1.0 + 3 * 0.044715 * xf * xf)
    dx = (dy.float() * dgelu)
    db = dx.reshape(-1, x.shape[-1]).sum(0) if bias is not None else None
    return dx.to(x.dtype), db


def attention_fwd(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                  causal: bool, scale: Optional[float] = None
                  ) -> Tuple[torch.Tensor, torch.Tensor]:
    """q,k,v: [B, H, S, D] -> (o [B,H,S,D], lse [B,H,S]) computed in fp32."""
    qf, kf, vf = q.float(), k.float(), v.float()
    D = q.shape[-1]
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        Sq, Sk = s.shape[-2], s.shape[-1]
        mask = torch.ones(Sq, Sk, dtype=torch.bool, device=s.device).tril(Sk - Sq)
        s = s.masked_fill(~mask, float("-inf"))
    lse = torch.logsumexp(s, dim=-1)
    p = torch.exp(s - lse[..., None])
    o = torch.matmul(p, vf)
    return o.to(q.dtype), lse


def attention_bwd(do: torch.Tensor, q: torch.Tensor, k: torch.Tensor,
                  v: torch.Tensor, o: torch.Tensor, lse: torch.Tensor,
                  causal: bool, scale: Optional[float] = None
                  ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    qf, kf, vf, dof, of = q.float(), k.float(), v.float(), do.float(), o.float()
    D = q.shape[-1]
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
    if causal:
        Sq, Sk = s.shape[-2], s.shape[-1]
        mask = torch.ones(Sq, Sk, dtype=torch.bool, device=s.device).tril(Sk - Sq)
        s = s.masked_fill(~mask, float("-inf"))
    p = torch.exp(s - lse[..., None].float())
    dv = torch.matmul(p.transpose(-1, -2), dof)
    dp = torch.matmul(dof, vf.transpose(-1, -2))
    delta = (dof * of).sum(dim=-1, keepdim=True)
    ds = p * (dp - delta) * scale
    dq = torch.matmul(ds, kf)
    dk = torch.matmul(ds.transpose(-1, -2), qf)
    return dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype)


def softmax_causal_fwd(scores: torch.Tensor, scale: float) -> torch.Tensor:
    """Fused scale + causal-mask + softmax on [B, H, Sq, Sk] scores (fp32 math)."""
    s = scores.float() * scale
    Sq, Sk = s.shape[-2], s.shape[-1]
    mask = torch.ones(Sq, Sk, dtype=torch.bool, device=s.device).tril(Sk - Sq)
    s = s.masked_fill(~mask, float("-inf"))
    return torch.softmax(s, dim=-1).to(scores.dtype)


def softmax_causal_bwd(dy: torch.Tensor, y: torch.Tensor, scale: float) -> torch.Tensor:
    yf, dyf = y.float(), dy.float()
    ds = yf * (dyf - (dyf * yf).sum(dim=-1, keepdim=True)) * scale
    return ds.to(y.dtype)


def cross_entropy_fwd(logits: torch.Tensor, labels: torch.Tensor,
                      ignore_index: int = -100
                      ) -> Tuple[torch.Tensor, torch.Tensor]:
    """logits [N, V] (any float dtype), labels [N] -> (loss [N] fp32, lse [N] fp32).

    loss = lse - logit[label]; 0 where ignored.
    """
    lf = logits.float()
    lse = torch.logsumexp(lf, dim=-1)
    valid = labels != ignore_index
    safe = labels.clamp(min=0)
    picked = lf.gather(1, safe[:, None]).squeeze(1)
    loss = torch.where(valid, lse - picked, torch.zeros_like(lse))
    return loss, lse


def cross_entropy_bwd(dloss: torch.Tensor, logits: torch.Tensor,
                      labels: torch.Tensor, lse: torch.Tensor,
                      ignore_index: int = -100) -> torch.Tensor:
    lf = logits.float()
    p = torch.exp(lf - lse[:, None])
    valid = (labels != ignore_index)
    safe = labels.clamp(min=0)
    onehot = torch.zeros_like(lf)
    onehot.scatter_(1, safe[:, None], 1.0)
    g = (p - onehot) * dloss[:, None] * valid[:, None]
    return g.to(logits.dtype)


def adamw_step(master: torch.Tensor, grad: torch.Tensor, exp_avg: torch.Tensor,
               exp_avg_sq: torch.Tensor, model_out: Optional[torch.Tensor],
               lr: float, beta1: float, beta2: float, eps: float,
               weight_decay: float, step: int) -> None:
    """Flat-buffer AdamW, fp32 master + bf16 model copy. In-place."""
    gf = grad.float()
    exp_avg.mul_(beta1).add_(gf, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    denom = (exp_avg_sq / bc2).sqrt().add_(eps)
    master.mul_(1 - lr * weight_decay)
    master.addcdiv_(exp_avg / bc1, denom, value=-lr)
    if model_out is not None:
        model_out.copy_(master.to(model_out.dtype))


def rope_fwd(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """x: [B, H, S, D]; cos/sin: [S, D/2] fp32. Interleaved-pair rotation."""
    xf = x.float()
    x1 = xf[..., 0::2]
    x2 = xf[..., 1::2]
    c = cos[None, None, :, :]
    s = sin[None, None, :, :]
    o1 = x1 * c - x2 * s
    o2 = x2 * c + x1 * s
    out = torch.stack((o1, o2), dim=-1).flatten(-2)
    return out.to(x.dtype)


def rope_bwd(dy: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    return rope_fwd(dy, cos, -sin)


def topp_sampling(probs: torch.Tensor, top_p: torch.Tensor,
                  seed: int = -1) -> Tuple[torch.Tensor, torch.Tensor]:
    """Nucleus sampling. probs [B, V] (rows sum to 1), top_p [B].

    Returns (ids [B,1] int64, probs_of_ids [B,1]). Reference semantics of
    ppfleetx/ops/topp_sampling.cu:377 (smallest prefix of sorted probs with
    cumsum >= p, sampled proportionally).
    """
    if seed >= 0:
        g = torch.Generator(device=probs.device)
        g.manual_seed(seed)
    else:
        g = None
    sorted_p, sorted_idx = torch.sort(probs.float(), dim=-1, descending=True)
    cum = torch.cumsum(sorted_p, dim=-1)
    # keep tokens until cumulative prob reaches top_p (always keep first)
    keep = cum - sorted_p < top_p[:, None]
    keep[:, 0] = True
    filt = sorted_p * keep
    filt = filt / filt.sum(dim=-1, keepdim=True)
    pick = torch.multinomial(filt, 1, generator=g)
    ids = sorted_idx.gather(1, pick)
    pp = probs.gather(1, ids)
    return ids, pp
