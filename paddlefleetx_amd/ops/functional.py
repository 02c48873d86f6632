"""Autograd wrappers over the HIP kernels (CUDA) / torch references (CPU).

Kernel parity map (reference -> here):
  paddle flash_attention (hybrid_model.py:284-301)  -> flash_attention
  incubate.softmax_mask_fuse_upper_triangle (:325)  -> fused_softmax_causal
  fused LayerNorm (paddle nn.LayerNorm fused path)  -> layernorm
  ParallelCrossEntropy inner kernel (:951)          -> cross_entropy(+parallel in parallel/tp.py)
  FusedAdamW (optims/optimizer.py:31)               -> fused_adamw_flat
  fused_gemm_epilogue gelu path                     -> bias_gelu
  ppfleetx/ops/topp_sampling.cu:377                 -> topp_sampling
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch

from paddlefleetx_amd.ops import _reference as ref
from paddlefleetx_amd.ops import hip_ext, use_hip

__all__ = [
    "layernorm", "layernorm_residual", "rmsnorm", "FusedLayerNorm",
    "FusedRMSNorm", "bias_gelu",
    "flash_attention", "flash_attention_packed", "fused_softmax_causal",
    "fused_softmax_bias", "cross_entropy",
    "fused_adamw_flat", "rope", "topp_sampling",
]


# ---------------------------------------------------------------------------
# LayerNorm / RMSNorm
# ---------------------------------------------------------------------------

class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        shape = x.shape
        x2 = x.contiguous().view(-1, shape[-1])
        if use_hip(x2):
            y, mean, rstd = hip_ext().layernorm_fwd(x2, weight, bias, eps)
        else:
            y, mean, rstd = ref.layernorm_fwd(x2, weight, bias, eps)
        ctx.save_for_backward(x2, weight, mean, rstd)
        return y.view(shape)

    @staticmethod
    def backward(ctx, dy):
        x2, weight, mean, rstd = ctx.saved_tensors
        dy2 = dy.contiguous().view(-1, dy.shape[-1])
        if use_hip(x2):
            dx, dw, db = hip_ext().layernorm_bwd(dy2, x2, weight, mean, rstd)
        else:
            dx, dw, db = ref.layernorm_bwd(dy2, x2, weight, mean, rstd)
        return dx.view(dy.shape), dw.to(weight.dtype), db.to(weight.dtype), None


def layernorm(x, weight, bias, eps: float = 1e-5):
    return _LayerNormFn.apply(x, weight, bias, eps)


class _LayerNormResidualFn(torch.autograd.Function):
    """y = LN(x + res); also returns s = x + res for the residual stream.
    One kernel does the add + stats + normalize (fwd) and the residual
    gradient join (bwd) — removes two full-tensor elementwise passes per
    call vs separate add + LN."""

    @staticmethod
    def forward(ctx, x, res, weight, bias, eps):
        shape = x.shape
        x2 = x.contiguous().view(-1, shape[-1])
        r2 = res.contiguous().view(-1, shape[-1])
        if use_hip(x2):
            y, mean, rstd, s = hip_ext().layernorm_fwd_residual(
                x2, r2, weight, bias, eps)
        else:
            y, mean, rstd, s = ref.layernorm_fwd_residual(
                x2, r2, weight, bias, eps)
        ctx.save_for_backward(s, weight, mean, rstd)
        return y.view(shape), s.view(shape)

    @staticmethod
    def backward(ctx, dy, ds):
        s2, weight, mean, rstd = ctx.saved_tensors
        dy2 = dy.contiguous().view(-1, dy.shape[-1])
        if ds is not None:
            ds2 = ds.contiguous().view(-1, dy.shape[-1])
            if use_hip(s2):
                dx, dw, db = hip_ext().layernorm_bwd_residual(
                    dy2, s2, weight, mean, rstd, ds2)
            else:
                dx, dw, db = ref.layernorm_bwd_residual(
                    dy2, s2, weight, mean, rstd, ds2)
        else:
            if use_hip(s2):
                dx, dw, db = hip_ext().layernorm_bwd(dy2, s2, weight, mean,
                                                     rstd)
            else:
                dx, dw, db = ref.layernorm_bwd(dy2, s2, weight, mean, rstd)
        dxv = dx.view(dy.shape)
        # d(x) = d(res) = dx (s = x + res is linear in both)
        return dxv, dxv, dw.to(weight.dtype), db.to(weight.dtype), None


def layernorm_residual(x, res, weight, bias, eps: float = 1e-5):
    """Fused (LN(x+res), x+res)."""
    return _LayerNormResidualFn.apply(x, res, weight, bias, eps)


class _RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        shape = x.shape
        x2 = x.contiguous().view(-1, shape[-1])
        if use_hip(x2):
            y, rstd = hip_ext().rmsnorm_fwd(x2, weight, eps)
        else:
            y, rstd = ref.rmsnorm_fwd(x2, weight, eps)
        ctx.save_for_backward(x2, weight, rstd)
        return y.view(shape)

    @staticmethod
    def backward(ctx, dy):
        x2, weight, rstd = ctx.saved_tensors
        dy2 = dy.contiguous().view(-1, dy.shape[-1])
        if use_hip(x2):
            dx, dw = hip_ext().rmsnorm_bwd(dy2, x2, weight, rstd)
        else:
            dx, dw = ref.rmsnorm_bwd(dy2, x2, weight, rstd)
        return dx.view(dy.shape), dw.to(weight.dtype), None


def rmsnorm(x, weight, eps: float = 1e-6):
    return _RMSNormFn.apply(x, weight, eps)


class FusedLayerNorm(torch.nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-5,
                 dtype: Optional[torch.dtype] = None):
        super().__init__()
        self.eps = eps
        self.weight = torch.nn.Parameter(torch.ones(hidden_size, dtype=dtype))
        self.bias = torch.nn.Parameter(torch.zeros(hidden_size, dtype=dtype))

    def forward(self, x):
        return layernorm(x, self.weight, self.bias, self.eps)


class FusedRMSNorm(torch.nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-6,
                 dtype: Optional[torch.dtype] = None):
        super().__init__()
        self.eps = eps
        self.weight = torch.nn.Parameter(torch.ones(hidden_size, dtype=dtype))

    def forward(self, x):
        return rmsnorm(x, self.weight, self.eps)


# ---------------------------------------------------------------------------
# Bias + GeLU (tanh approx) fusion
# ---------------------------------------------------------------------------

class _BiasGeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        xc = x.contiguous()
        if use_hip(xc):
            y = hip_ext().bias_gelu_fwd(xc.view(-1, xc.shape[-1]),
                                        bias if bias is not None else torch.Tensor())
            y = y.view(xc.shape)
        else:
            y = ref.bias_gelu_fwd(xc, bias)
        ctx.save_for_backward(xc, bias if bias is not None else None)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        if ctx.has_bias:
            x, bias = ctx.saved_tensors
        else:
            (x,) = ctx.saved_tensors
            bias = None
        dyc = dy.contiguous()
        if use_hip(x):
            dx, db = hip_ext().bias_gelu_bwd(dyc.view(-1, x.shape[-1]),
                                             x.view(-1, x.shape[-1]),
                                             bias if bias is not None else torch.Tensor())
            dx = dx.view(x.shape)
            db = db if ctx.has_bias else None
        else:
            dx, db = ref.bias_gelu_bwd(dyc, x, bias)
        if db is not None and bias is not None:
            db = db.to(bias.dtype)
        return dx, db


def bias_gelu(x, bias=None):
    return _BiasGeluFn.apply(x, bias)


# ---------------------------------------------------------------------------
# Flash attention (causal), bf16, [B, H, S, D]
# ---------------------------------------------------------------------------

class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, scale, p_drop, seed):
        scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        if use_hip(q):
            o, lse = hip_ext().attn_fwd(q, k, v, causal, scale,
                                        p_drop, seed)
        else:
            # CPU reference has no Philox twin; dropout handled by the
            # caller's torch path there
            assert p_drop == 0.0, "CPU path: dropout not fused"
            o, lse = ref.attention_fwd(q, k, v, causal, scale)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal, ctx.scale = causal, scale
        ctx.p_drop, ctx.seed = p_drop, seed
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        do = do.contiguous()
        if use_hip(q):
            dq, dk, dv = hip_ext().attn_bwd(do, q, k, v, o, lse,
                                            ctx.causal, ctx.scale,
                                            ctx.p_drop, ctx.seed)
        else:
            dq, dk, dv = ref.attention_bwd(do, q, k, v, o, lse,
                                           ctx.causal, ctx.scale)
        return dq, dk, dv, None, None, None, None


def _drop_seed() -> int:
    # drawn from the torch CPU generator so the per-rank local_seed
    # discipline (parallel/rng.py model_parallel_rng) governs the mask
    return int(torch.randint(0, 2 ** 62, (1,)).item())


def flash_attention(q, k, v, causal: bool = True,
                    scale: Optional[float] = None, p_drop: float = 0.0):
    """q,k,v: [B, H, S, D] -> o: [B, H, S, D]. O(S) memory, online
    softmax; in-kernel Philox attention dropout when p_drop > 0."""
    seed = _drop_seed() if p_drop > 0 else 0
    return _FlashAttnFn.apply(q, k, v, causal, scale, p_drop, seed)


class _FlashAttnPackedFn(torch.autograd.Function):
    """Packed-QKV flash attention: input [B, S, h, 3, D] (fused-QKV linear
    output viewed), output [B, S, h*D] — no split/transpose/cat copies."""

    @staticmethod
    def forward(ctx, qkv, num_heads, scale, p_drop, seed):
        scale = scale if scale is not None else 1.0 / math.sqrt(qkv.shape[-1])
        if use_hip(qkv):
            o, lse = hip_ext().attn_fwd_packed(qkv, num_heads, scale,
                                               p_drop, seed)
        else:
            B, S, h, _, D = qkv.shape
            q = qkv[:, :, :, 0].permute(0, 2, 1, 3)
            k = qkv[:, :, :, 1].permute(0, 2, 1, 3)
            v = qkv[:, :, :, 2].permute(0, 2, 1, 3)
            o4, lse = ref.attention_fwd(q, k, v, True, scale)
            o = o4.permute(0, 2, 1, 3).reshape(B, S, h * D)
        ctx.save_for_backward(qkv, o, lse)
        ctx.num_heads, ctx.scale = num_heads, scale
        ctx.p_drop, ctx.seed = p_drop, seed
        return o

    @staticmethod
    def backward(ctx, do):
        qkv, o, lse = ctx.saved_tensors
        if use_hip(qkv):
            dqkv = hip_ext().attn_bwd_packed(do.contiguous(), qkv, o, lse,
                                             ctx.num_heads, ctx.scale,
                                             ctx.p_drop, ctx.seed)
        else:
            B, S, h, _, D = qkv.shape
            q = qkv[:, :, :, 0].permute(0, 2, 1, 3)
            k = qkv[:, :, :, 1].permute(0, 2, 1, 3)
            v = qkv[:, :, :, 2].permute(0, 2, 1, 3)
            o4 = o.view(B, S, h, D).permute(0, 2, 1, 3)
            do4 = do.view(B, S, h, D).permute(0, 2, 1, 3)
            dq, dk, dv = ref.attention_bwd(do4, q, k, v, o4, lse, True,
                                           ctx.scale)
            dqkv = torch.stack(
                (dq.permute(0, 2, 1, 3), dk.permute(0, 2, 1, 3),
                 dv.permute(0, 2, 1, 3)), dim=3)
        return dqkv, None, None, None, None


def flash_attention_packed(qkv, num_heads: int,
                           scale: Optional[float] = None,
                           p_drop: float = 0.0):
    """qkv [B, S, h, 3, D] -> o [B, S, h*D]; causal; in-kernel Philox
    attention dropout when p_drop > 0 (GPU only)."""
    if p_drop > 0:
        assert qkv.is_cuda, "fused attention dropout is the GPU path"
    seed = _drop_seed() if p_drop > 0 else 0
    return _FlashAttnPackedFn.apply(qkv.contiguous(), num_heads, scale,
                                    p_drop, seed)


# ---------------------------------------------------------------------------
# Fused scale + causal mask + softmax (non-flash core_attn path)
# ---------------------------------------------------------------------------

class _SoftmaxCausalFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, scores, scale):
        sc = scores.contiguous()
        if use_hip(sc):
            y = hip_ext().softmax_causal_fwd(sc, scale)
        else:
            y = ref.softmax_causal_fwd(sc, scale)
        ctx.save_for_backward(y)
        ctx.scale = scale
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        dy = dy.contiguous()
        if use_hip(y):
            ds = hip_ext().softmax_causal_bwd(dy, y, ctx.scale)
        else:
            ds = ref.softmax_causal_bwd(dy, y, ctx.scale)
        return ds, None


def fused_softmax_causal(scores, scale: float = 1.0):
    """[B,H,Sq,Sk] scores -> causal softmax(scale*scores). Fused mask, no mask tensor."""
    return _SoftmaxCausalFn.apply(scores, scale)


def _bias_period(s_shape, b_shape):
    """For bias broadcasting over one contiguous run of leading dims
    (e.g. logits [B, S, h, Q, K] vs bias [B, 1, h, Q, K]) return
    (inner, outer) row periods for the fused kernel, or None."""
    if len(s_shape) != len(b_shape) or s_shape[-1] != b_shape[-1]:
        return None
    lead_s, lead_b = s_shape[:-1], b_shape[:-1]
    bc = [i for i, (a, b) in enumerate(zip(lead_s, lead_b))
          if b == 1 and a != 1]
    eq = all(a == b for i, (a, b) in enumerate(zip(lead_s, lead_b))
             if i not in bc)
    if not eq:
        return None
    if not bc:
        return 1, 1  # no broadcast: brow == row
    if bc != list(range(bc[0], bc[-1] + 1)):
        return None  # non-contiguous broadcast run
    inner = 1
    for d in lead_s[bc[-1] + 1:]:
        inner *= d
    outer = inner
    for d in lead_s[bc[0]:bc[-1] + 1]:
        outer *= d
    return inner, outer


class _SoftmaxBiasFn(torch.autograd.Function):
    """softmax(scores * scale + bias) over the last dim with the bias
    broadcast handled inside ONE kernel — the fused gated-attention core
    (reference fused_gate_attention, protein_folding/attentions.py:126)."""

    @staticmethod
    def forward(ctx, scores, bias, scale, period):
        sc = scores.contiguous()
        bc_ = bias.contiguous()
        y = hip_ext().softmax_bias_fwd(sc, bc_, period[0], period[1], scale)
        ctx.save_for_backward(y)
        ctx.scale = scale
        ctx.bias_shape = bias.shape
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        dy = dy.contiguous()
        ds = hip_ext().softmax_causal_bwd(dy, y, ctx.scale)  # generic bwd
        dlogits = ds if ctx.scale == 1.0 else ds / ctx.scale
        red = [i for i, (a, b) in enumerate(zip(y.shape, ctx.bias_shape))
               if b == 1 and a != 1]
        db = dlogits.sum(dim=red, keepdim=True) if red else dlogits
        return ds, db.to(y.dtype), None, None


def fused_softmax_bias(scores, bias, scale: float = 1.0):
    """Fused softmax(scale*scores + bias) when on GPU and the bias
    broadcast is a contiguous leading run; torch fallback otherwise."""
    if use_hip(scores):
        period = _bias_period(scores.shape, bias.shape)
        if period is not None:
            return _SoftmaxBiasFn.apply(scores, bias, scale, period)
    return torch.softmax(scores.float() * scale + bias.float(),
                         dim=-1).to(scores.dtype)


# ---------------------------------------------------------------------------
# Cross entropy (single-rank). Vocab-parallel version lives in parallel/tp.py
# and reuses these kernels plus RCCL allreduce.
# ---------------------------------------------------------------------------

class _CrossEntropyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, ignore_index):
        lc = logits.contiguous()
        if use_hip(lc):
            loss, lse = hip_ext().cross_entropy_fwd(lc, labels, ignore_index)
        else:
            loss, lse = ref.cross_entropy_fwd(lc, labels, ignore_index)
        ctx.save_for_backward(lc, labels, lse)
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, labels, lse = ctx.saved_tensors
        if use_hip(logits):
            dl = hip_ext().cross_entropy_bwd(dloss.contiguous(), logits, labels,
                                             lse, ctx.ignore_index)
        else:
            dl = ref.cross_entropy_bwd(dloss, logits, labels, lse, ctx.ignore_index)
        return dl, None, None


def cross_entropy(logits, labels, ignore_index: int = -100):
    """logits [N, V], labels [N] -> per-token loss [N] (fp32)."""
    return _CrossEntropyFn.apply(logits, labels, ignore_index)


# ---------------------------------------------------------------------------
# Fused AdamW over flat fp32 master buffers
# ---------------------------------------------------------------------------

def fused_adamw_flat(master: torch.Tensor, grad: torch.Tensor,
                     exp_avg: torch.Tensor, exp_avg_sq: torch.Tensor,
                     model_out: Optional[torch.Tensor], lr: float,
                     beta1: float, beta2: float, eps: float,
                     weight_decay: float, step: int) -> None:
    """In-place AdamW on a flat fp32 master buffer; writes bf16 model copy."""
    if use_hip(master):
        hip_ext().adamw_flat(master, grad, exp_avg, exp_avg_sq,
                             model_out if model_out is not None else torch.Tensor(),
                             lr, beta1, beta2, eps, weight_decay, step)
    else:
        ref.adamw_step(master, grad, exp_avg, exp_avg_sq, model_out,
                       lr, beta1, beta2, eps, weight_decay, step)


# ---------------------------------------------------------------------------
# RoPE
# ---------------------------------------------------------------------------

class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin):
        xc = x.contiguous()
        if use_hip(xc):
            y = hip_ext().rope_fwd(xc, cos, sin)
        else:
            y = ref.rope_fwd(xc, cos, sin)
        ctx.save_for_backward(cos, sin)
        return y

    @staticmethod
    def backward(ctx, dy):
        cos, sin = ctx.saved_tensors
        dyc = dy.contiguous()
        if use_hip(dyc):
            dx = hip_ext().rope_fwd(dyc, cos, -sin)
        else:
            dx = ref.rope_bwd(dyc, cos, sin)
        return dx, None, None


def rope(x, cos, sin):
    """Rotary embedding, interleaved pairs. x [B,H,S,D], cos/sin [S,D/2] fp32."""
    return _RopeFn.apply(x, cos, sin)


# ---------------------------------------------------------------------------
# Top-p (nucleus) sampling
# ---------------------------------------------------------------------------

def topp_sampling(probs: torch.Tensor, top_p: torch.Tensor, seed: int = -1
                  ) -> Tuple[torch.Tensor, torch.Tensor]:
    """probs [B, V] rows sum to 1; top_p [B] -> (ids [B,1], prob [B,1]).

    Replaces ppfleetx/ops/topp_sampling.cu (CUB radix sort + prefix scan) with
    a gfx950 kernel.
    """
    if use_hip(probs):
        # rocPRIM radix sort (torch.sort) + HIP scan/cutoff/draw kernel
        pf = probs.float().contiguous()
        sorted_p, sorted_idx = torch.sort(pf, dim=-1, descending=True)
        if seed >= 0:
            g = torch.Generator(device=probs.device)
            g.manual_seed(seed)
            u = torch.rand(probs.shape[0], device=probs.device, generator=g)
        else:
            u = torch.rand(probs.shape[0], device=probs.device)
        ids, pp = hip_ext().topp_select(sorted_p, sorted_idx,
                                        top_p.float().contiguous(),
                                        u.contiguous())
        return ids, pp.to(probs.dtype)
    return ref.topp_sampling(probs, top_p, seed)
