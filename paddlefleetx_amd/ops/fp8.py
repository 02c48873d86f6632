"""Opt-in fp8 (OCP e4m3) GEMM path for serving.

gfx950's MFMA fp8 rate is 2x bf16 (guide §4: OCP formats, not FNUZ);
hipBLASLt exposes it through torch._scaled_mm. This module converts
Linear-like layers to per-tensor-scaled fp8 weight storage with dynamic
per-tensor activation scaling — an inference-time path (the training
bench stays bf16 per the baseline contract).

Usage:  from paddlefleetx_amd.ops.fp8 import convert_fp8_linears
        convert_fp8_linears(model)           # after load, before eval
"""

from __future__ import annotations

from typing import Iterable, Optional

import torch
import torch.nn as nn

from paddlefleetx_amd.utils.log import logger

E4M3_MAX = 448.0


def fp8_available() -> bool:
    if not torch.cuda.is_available():
        return False
    try:
        a = torch.randn(16, 16, device="cuda")
        b = torch.randn(16, 16, device="cuda")
        sa = (a.abs().amax() / E4M3_MAX).clamp(min=1e-12)
        sb = (b.abs().amax() / E4M3_MAX).clamp(min=1e-12)
        torch._scaled_mm((a / sa).to(torch.float8_e4m3fn),
                         (b / sb).to(torch.float8_e4m3fn).t(),
                         scale_a=sa.float(), scale_b=sb.float(),
                         out_dtype=torch.bfloat16)
        return True
    except Exception as e:  # pragma: no cover - hardware/library dependent
        logger.warning(f"fp8 path unavailable: {e}")
        return False


class Fp8Linear(nn.Module):
    """y = x @ W^T (+ b) with fp8 storage + fp8 MFMA GEMM.

    Weight quantized once per-tensor at conversion; activations
    quantized dynamically per call (per-tensor absmax)."""

    def __init__(self, linear):
        super().__init__()
        w = linear.weight.data
        scale = (w.abs().amax().float() / E4M3_MAX).clamp(min=1e-12)
        self.register_buffer("wq", (w.float() / scale)
                             .to(torch.float8_e4m3fn))
        self.register_buffer("w_scale", scale)
        self.bias = getattr(linear, "bias", None)
        self.out_features, self.in_features = w.shape

    def forward(self, x):
        shape = x.shape
        x2 = x.reshape(-1, shape[-1])
        xs = (x2.abs().amax().float() / E4M3_MAX).clamp(min=1e-12)
        xq = (x2.float() / xs).to(torch.float8_e4m3fn)
        # _scaled_mm wants the second operand column-major
        y = torch._scaled_mm(xq, self.wq.t(), scale_a=xs,
                             scale_b=self.w_scale,
                             out_dtype=x.dtype if x.dtype in
                             (torch.bfloat16, torch.float16)
                             else torch.bfloat16)
        if self.bias is not None:
            y = y + self.bias
        return y.reshape(*shape[:-1], self.out_features).to(x.dtype)


_LINEAR_NAMES = ("Linear", "ColumnParallelLinear", "RowParallelLinear")


def convert_fp8_linears(model: nn.Module,
                        include: Optional[Iterable[str]] = None,
                        min_features: int = 1024) -> int:
    """Replace large Linear-like layers with Fp8Linear (serving path).
    TP linears are only converted at mp==1 (their comm wrappers live in
    forward, which this replaces)."""
    if not fp8_available():
        logger.warning("fp8 conversion skipped (no _scaled_mm support)")
        return 0
    count = 0
    for name, mod in list(model.named_modules()):
        for child_name, child in list(mod.named_children()):
            full = f"{name}.{child_name}" if name else child_name
            if type(child).__name__ not in _LINEAR_NAMES:
                continue
            if include and not any(k in full for k in include):
                continue
            w = getattr(child, "weight", None)
            if w is None or w.dim() != 2 or min(w.shape) < min_features:
                continue
            if type(child).__name__ != "Linear":
                from paddlefleetx_amd.parallel.env import get_hcg
                try:
                    if get_hcg().get_model_parallel_world_size() != 1:
                        continue
                except Exception:
                    pass
            setattr(mod, child_name, Fp8Linear(child))
            count += 1
    logger.info(f"converted {count} Linear layers to fp8 (e4m3) GEMMs")
    return count
