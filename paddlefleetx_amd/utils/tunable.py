"""hipBLASLt/rocBLAS GEMM algorithm selection via PyTorch TunableOp.

The reference leans on Paddle's pre-tuned GEMM dispatch; on ROCm the
equivalent is TunableOp: an offline per-shape algorithm search whose
winners we ship as a checked-in table (configs/tunableop_gfx950.csv,
tuned on MI355X for the GPT-6.7B hot shapes — measured 1.3-2.1 PF/s vs
1.0-1.5 for the default picks). At runtime the table is loaded
read-only; unknown shapes fall back to the default heuristic with zero
tuning cost.

Re-tune (on a GPU box) with:
  PYTORCH_TUNABLEOP_ENABLED=1 PYTORCH_TUNABLEOP_TUNING=1 \
  PYTORCH_TUNABLEOP_FILENAME=/tmp/tun.csv python bench.py --steps 3
then merge /tmp/tun0.csv into configs/tunableop_gfx950.csv.
"""

from __future__ import annotations

import os

import torch

_DEFAULT = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "configs", "tunableop_gfx950.csv")

_loaded = False


def enable_tuned_gemms(csv_path: str | None = None) -> bool:
    """Load the tuned GEMM table (read-only). No-op without a GPU, when
    the table is missing, or when the user drives TunableOp via its own
    PYTORCH_TUNABLEOP_* env (e.g. a re-tuning run)."""
    global _loaded
    if _loaded:
        return True
    if not torch.cuda.is_available():
        return False
    env = os.environ.get("PYTORCH_TUNABLEOP_ENABLED", "").strip()
    if env and env != "0":
        return False  # user-driven session (tuning or custom file)
    path = csv_path or _DEFAULT
    if not os.path.exists(path):
        return False
    import torch.cuda.tunable as tunable
    tunable.enable(True)
    tunable.tuning_enable(False)  # lookup only; never tune in production
    tunable.read_file(path)
    _loaded = True
    from paddlefleetx_amd.utils.log import logger
    logger.info(f"TunableOp: loaded tuned GEMM table {path}")
    return True
