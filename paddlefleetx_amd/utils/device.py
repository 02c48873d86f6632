"""Device helpers (reference ppfleetx/utils/device.py:44 synchronize).

MI355X-native: one process per GPU; LOCAL_RANK selects the device.
"""

from __future__ import annotations

import os

import torch


def get_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", 0))


def set_device() -> torch.device:
    """Bind this process to its LOCAL_RANK GPU (torchrun layout)."""
    if torch.cuda.is_available():
        dev = torch.device(f"cuda:{get_local_rank() % torch.cuda.device_count()}")
        torch.cuda.set_device(dev)
        return dev
    return torch.device("cpu")


def synchronize() -> None:
    """Device sync (reference device.py:44) — used around timing and
    checkpoint writes."""
    if torch.cuda.is_available():
        torch.cuda.synchronize()


def device_memory_stats() -> dict:
    if not torch.cuda.is_available():
        return {}
    return {
        "allocated_gb": torch.cuda.memory_allocated() / 2 ** 30,
        "reserved_gb": torch.cuda.memory_reserved() / 2 ** 30,
        "peak_gb": torch.cuda.max_memory_allocated() / 2 ** 30,
    }
