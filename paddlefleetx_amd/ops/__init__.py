"""MI355X-native fused ops.

Every hot op has (a) a hand-written gfx950 HIP kernel in csrc/ compiled into
the in-tree extension `_fleetx_hip`, and (b) a plain-PyTorch fp32 reference
(`_reference.py`) used on CPU and as the numerics twin in tests.

On a CUDA(ROCm) device the HIP extension is REQUIRED: ops raise if it is
missing rather than silently falling back to eager.
"""

from __future__ import annotations

import os

import torch

_EXT = None
_EXT_ERR = None


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from paddlefleetx_amd.ops import _fleetx_hip  # in-tree built .so
        _EXT = _fleetx_hip
    except ImportError as e:  # pragma: no cover - GPU box only
        _EXT_ERR = e
    return _EXT


def hip_ext():
    """Return the HIP extension module, raising loudly if unavailable."""
    ext = _load_extension()
    if ext is None:
        raise RuntimeError(
            "paddlefleetx_amd HIP extension (_fleetx_hip) is not built. "
            "Run `python setup.py build_ext --inplace` (gfx950). "
            f"Import error: {_EXT_ERR}")
    return ext


def has_hip_ext() -> bool:
    return _load_extension() is not None


def use_hip(t: torch.Tensor) -> bool:
    """HIP kernels run for CUDA tensors; CPU tensors use the reference path."""
    return t.is_cuda


from paddlefleetx_amd.ops.functional import (  # noqa: E402,F401
    FusedLayerNorm,
    FusedRMSNorm,
    bias_gelu,
    cross_entropy,
    flash_attention,
    flash_attention_packed,
    fused_adamw_flat,
    fused_softmax_bias,
    fused_softmax_causal,
    layernorm,
    layernorm_residual,
    rmsnorm,
    rope,
    topp_sampling,
)
