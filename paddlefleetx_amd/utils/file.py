"""Filesystem helpers (reference ppfleetx/utils/file.py + download.py).

This environment has no network: `download()` only resolves local paths /
caches and raises a clear error for URLs.
"""

from __future__ import annotations

import hashlib
import os
import shutil
from typing import Optional

from paddlefleetx_amd.utils.log import logger


def mkdir_if_not_exist(path: str) -> str:
    os.makedirs(path, exist_ok=True)
    return path


def is_url(path: str) -> bool:
    return path.startswith(("http://", "https://"))


def md5file(path: str) -> str:
    h = hashlib.md5()
    with open(path, "rb") as f:
        for chunk in iter(lambda: f.read(1 << 20), b""):
            h.update(chunk)
    return h.hexdigest()


def download(url_or_path: str, target_dir: str = "./cache",
             md5sum: Optional[str] = None) -> str:
    """Local-path resolver (reference utils/download.py surface). URLs are
    refused: this deployment is offline — stage files onto the node."""
    if is_url(url_or_path):
        raise RuntimeError(
            f"cannot download {url_or_path}: no network access in this "
            "environment. Copy the file to local storage and pass its path.")
    if not os.path.exists(url_or_path):
        raise FileNotFoundError(url_or_path)
    if md5sum and md5file(url_or_path) != md5sum:
        raise IOError(f"md5 mismatch for {url_or_path}")
    return url_or_path


def copy_to(src: str, dst_dir: str) -> str:
    mkdir_if_not_exist(dst_dir)
    dst = os.path.join(dst_dir, os.path.basename(src))
    shutil.copy2(src, dst)
    logger.info(f"copied {src} -> {dst}")
    return dst
