"""Model/data artifact resolution (reference ppfleetx/utils/download.py).

This environment has NO network: `cached_path` resolves local paths and
well-known cache directories and raises a clear error instead of
attempting a download.
"""

from __future__ import annotations

import os
from typing import Optional

SEARCH_DIRS = [
    os.environ.get("PFX_CACHE_DIR", ""),
    os.path.expanduser("~/.cache/paddlefleetx_amd"),
    "/root/data",
]


def cached_path(name_or_path: str, subdir: Optional[str] = None) -> str:
    """Return an existing local path for `name_or_path`, searching the
    cache dirs; raise FileNotFoundError with download guidance (offline:
    the reference would fetch from BOS/HF here)."""
    if os.path.exists(name_or_path):
        return name_or_path
    for base in SEARCH_DIRS:
        if not base:
            continue
        cand = os.path.join(base, subdir or "", name_or_path)
        if os.path.exists(cand):
            return cand
    raise FileNotFoundError(
        f"artifact '{name_or_path}' not found locally (searched "
        f"{[b for b in SEARCH_DIRS if b]}); this environment has no "
        "network — place the file in one of those directories or set "
        "PFX_CACHE_DIR")
