"""Version gates (reference ppfleetx/utils/version.py + the
version_check gate in eager_engine.py:96)."""

from __future__ import annotations

__version__ = "0.1.0"

MIN_TORCH = (2, 4)
MIN_ROCM = (6, 0)


def version_check() -> None:
    import torch
    parts = torch.__version__.split("+")[0].split(".")
    tv = tuple(int(x) for x in parts[:2])
    if tv < MIN_TORCH:
        raise RuntimeError(
            f"paddlefleetx_amd needs torch>={'.'.join(map(str, MIN_TORCH))} "
            f"(ROCm build), found {torch.__version__}")
    hip = getattr(torch.version, "hip", None)
    if torch.cuda.is_available() and not hip:
        raise RuntimeError("a ROCm/HIP torch build is required "
                           "(CUDA builds are not supported)")
