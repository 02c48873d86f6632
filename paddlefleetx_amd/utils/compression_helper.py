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


class QuantizedLinear(nn.Module):
    """Int8 weight-only quantized Linear (per-out-channel scales)."""

    def __init__(self, linear: nn.Linear):
        super().__init__()
        w = linear.weight.data.float()
        scale = w.abs().amax(dim=1, keepdim=True).clamp(min=1e-8) / 127.0
        self.register_buffer("qweight", torch.round(w / scale).to(torch.int8))
        self.register_buffer("scale", scale.to(torch.float32))
        self.bias = linear.bias
        self.in_features = linear.in_features
        self.out_features = linear.out_features

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
            if isinstance(child, nn.Linear) and \
                    (not include or any(k in full for k in include)):
                setattr(mod, child_name, QuantizedLinear(child))
                count += 1
    logger.info(f"quantized {count} Linear layers to int8 weights")
    return count


def quantization_error(model_fp: nn.Module, model_q: nn.Module,
                       sample: Tuple[torch.Tensor, ...]) -> float:
    with torch.no_grad():
        a = model_fp(*sample)
        b = model_q(*sample)
    a = a[0] if isinstance(a, tuple) else a
    b = b[0] if isinstance(b, tuple) else b
    return float((a.float() - b.float()).abs().mean())
