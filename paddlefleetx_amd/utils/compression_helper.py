"""Model compression: pruning + quantization.

Reference: ppfleetx/utils/compression_helper.py — prune_model :175
(paddleslim structured prune) and quant_model :210 (QAT wrapper), driven
by engine.compress_model (eager_engine.py:757-774).

MI355X-native: magnitude pruning applies masks in place (the HIP GEMMs
run dense — pruning is a model-size/regularization tool here; 2:1
structured sparsity MFMA is a future fp8-path optimization), and
quantization packs Linear weights to int8 with per-channel scales for
inference export.
"""

from __future__ import annotations

from typing import Dict, Iterable, List, Optional, Tuple

import torch
import torch.nn as nn

from paddlefleetx_amd.utils.log import logger


def prune_model(model: nn.Module, ratio: float = 0.125,
                structured: bool = True,
                include: Optional[Iterable[str]] = None) -> Dict[str, float]:
    """Magnitude-prune Linear weights in place; returns per-layer sparsity.

    structured=True removes whole output channels (rows); otherwise
    element-wise unstructured masking.
    """
    report = {}
    for name, mod in model.named_modules():
        if not isinstance(mod, nn.Linear):
            continue
        if include and not any(k in name for k in include):
            continue
        w = mod.weight.data
        if structured:
            norms = w.float().abs().mean(dim=1)
            k = int(w.shape[0] * ratio)
            if k == 0:
                continue
            idx = torch.topk(norms, k, largest=False).indices
            w[idx] = 0
            if mod.bias is not None:
                mod.bias.data[idx] = 0
        else:
            k = int(w.numel() * ratio)
            if k == 0:
                continue
            thresh = torch.kthvalue(w.float().abs().reshape(-1), k).values
            w[w.abs() <= thresh] = 0
        report[name] = float((w == 0).float().mean())
    logger.info(f"pruned {len(report)} Linear layers (ratio {ratio}, "
                f"structured={structured})")
    return report


_LINEAR_CLASS_NAMES = ("Linear", "ColumnParallelLinear",
                       "RowParallelLinear", "ColumnSequenceParallelLinear",
                       "RowSequenceParallelLinear")


def _is_linear_like(mod: nn.Module) -> bool:
    """nn.Linear or a TP linear that degenerates to one at mp==1
    (weight [out, in] + optional bias; the reference's paddleslim QAT is
    likewise a single-card training feature)."""
    if isinstance(mod, nn.Linear):
        return True
    if type(mod).__name__ in _LINEAR_CLASS_NAMES[1:]:
        from paddlefleetx_amd.parallel.env import get_hcg
        try:
            return get_hcg().get_model_parallel_world_size() == 1
        except Exception:
            return True
    return False


class QuantizedLinear(nn.Module):
    """Int8 weight-only quantized Linear (per-out-channel scales)."""

    def __init__(self, linear):
        super().__init__()
        w = linear.weight.data.float()
        scale = w.abs().amax(dim=1, keepdim=True).clamp(min=1e-8) / 127.0
        self.register_buffer("qweight", torch.round(w / scale).to(torch.int8))
        self.register_buffer("scale", scale.to(torch.float32))
        self.bias = getattr(linear, "bias", None)
        self.out_features, self.in_features = linear.weight.shape

    def forward(self, x):
        w = (self.qweight.float() * self.scale).to(x.dtype)
        return torch.nn.functional.linear(x, w, self.bias)


def quant_model(model: nn.Module,
                include: Optional[Iterable[str]] = None) -> int:
    """Replace Linear layers with int8 weight-only QuantizedLinear."""
    count = 0
    for name, mod in list(model.named_modules()):
        for child_name, child in list(mod.named_children()):
            full = f"{name}.{child_name}" if name else child_name
            if _is_linear_like(child) and \
                    (not include or any(k in full for k in include)):
                setattr(mod, child_name, QuantizedLinear(child))
                count += 1
    logger.info(f"quantized {count} Linear layers to int8 weights")
    return count


class _FakeQuant(torch.autograd.Function):
    """Straight-through fake quantization: round-to-grid forward,
    identity gradient (the paddleslim QAT training-time op)."""

    @staticmethod
    def forward(ctx, x, scale, qmin, qmax):
        q = torch.clamp(torch.round(x / scale), qmin, qmax)
        return q * scale

    @staticmethod
    def backward(ctx, dy):
        return dy, None, None, None


def fake_quant(x, scale, bits: int = 8):
    qmax = 2 ** (bits - 1) - 1
    return _FakeQuant.apply(x, scale, -qmax - 1, qmax)


class QATLinear(nn.Module):
    """Training-time fake-quant Linear (QAT): per-out-channel int8 weight
    fake-quant + per-tensor activation fake-quant with a moving-average
    absmax observer (reference quant_model:
    paddleslim.dygraph.quant.QAT semantics, compression_helper.py:210)."""

    def __init__(self, linear, bits: int = 8,
                 act_momentum: float = 0.9):
        super().__init__()
        self.weight = linear.weight
        self.bias = getattr(linear, "bias", None)
        self.bits = bits
        self.act_momentum = act_momentum
        self.register_buffer("act_absmax", torch.zeros(()))
        self.out_features, self.in_features = linear.weight.shape

    def _weight_scale(self):
        qmax = 2 ** (self.bits - 1) - 1
        return self.weight.detach().float().abs().amax(
            dim=1, keepdim=True).clamp(min=1e-8) / qmax

    def forward(self, x):
        qmax = 2 ** (self.bits - 1) - 1
        if self.training:
            cur = x.detach().float().abs().amax()
            if float(self.act_absmax) == 0.0:
                self.act_absmax.fill_(cur)
            else:
                self.act_absmax.mul_(self.act_momentum).add_(
                    cur * (1 - self.act_momentum))
        a_scale = (self.act_absmax.clamp(min=1e-8) / qmax).to(x.dtype)
        xq = fake_quant(x, a_scale, self.bits)
        wq = fake_quant(self.weight,
                        self._weight_scale().to(self.weight.dtype),
                        self.bits)
        return torch.nn.functional.linear(xq, wq, self.bias)

    def to_inference(self) -> "QuantizedLinear":
        lin = nn.Linear(self.in_features, self.out_features,
                        bias=self.bias is not None)
        lin.weight = self.weight
        if self.bias is not None:
            lin.bias = self.bias
        return QuantizedLinear(lin)


def qat_model(model: nn.Module, include: Optional[Iterable[str]] = None,
              bits: int = 8) -> int:
    """Wrap Linear layers for quantization-aware training (fake-quant in
    the forward, full-precision master weights + STE gradients)."""
    count = 0
    for name, mod in list(model.named_modules()):
        for child_name, child in list(mod.named_children()):
            full = f"{name}.{child_name}" if name else child_name
            if _is_linear_like(child) and \
                    (not include or any(k in full for k in include)):
                setattr(mod, child_name, QATLinear(child, bits=bits))
                count += 1
    logger.info(f"QAT-wrapped {count} Linear layers ({bits}-bit fake quant)")
    return count


def convert_qat(model: nn.Module) -> int:
    """Fold trained QAT layers to int8 weight-only QuantizedLinear for
    export (the reference's quanter.quantize->save path)."""
    count = 0
    for name, mod in list(model.named_modules()):
        for child_name, child in list(mod.named_children()):
            if isinstance(child, QATLinear):
                setattr(mod, child_name, child.to_inference())
                count += 1
    logger.info(f"converted {count} QAT layers to int8")
    return count


def quantization_error(model_fp: nn.Module, model_q: nn.Module,
                       sample: Tuple[torch.Tensor, ...]) -> float:
    with torch.no_grad():
        a = model_fp(*sample)
        b = model_q(*sample)
    a = a[0] if isinstance(a, tuple) else a
    b = b[0] if isinstance(b, tuple) else b
    return float((a.float() - b.float()).abs().mean())
