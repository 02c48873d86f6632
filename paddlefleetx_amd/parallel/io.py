"""Engine-independent distributed checkpoint save/load.

Reference: ppfleetx/distributed/apis/io.py:28-151 — mirrors EagerEngine's
layout (`epoch_X_step_Y/mp_XX_sharding_XX_pp_XX/`), only dp_rank 0 writes
(io.py:44-46).
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from paddlefleetx_amd.parallel.env import get_hcg
from paddlefleetx_amd.utils.log import logger


def _subdir() -> str:
    hcg = get_hcg()
    return "mp_{:02d}_sharding_{:02d}_pp_{:02d}".format(
        hcg.get_model_parallel_rank(), hcg.get_sharding_parallel_rank(),
        hcg.get_pipe_parallel_rank())


def save(output_dir: str, model: torch.nn.Module, optimizer=None,
         epoch: int = 0, step: int = 0, extra: Optional[dict] = None) -> Optional[str]:
    hcg = get_hcg()
    if hcg.get_data_parallel_rank() != 0:
        return None
    out = os.path.join(output_dir, f"epoch_{epoch}_step_{step}", _subdir())
    os.makedirs(out, exist_ok=True)
    torch.save(model.state_dict(), os.path.join(out, "model.pdparams"))
    if optimizer is not None:
        torch.save(optimizer.state_dict(),
                   os.path.join(out, "model_state.pdopt"))
    meta = {"epoch": epoch, "step": step,
            "cpu_rng_state": torch.get_rng_state()}
    if torch.cuda.is_available():
        meta["cuda_rng_state"] = torch.cuda.get_rng_state()
    meta.update(extra or {})
    torch.save(meta, os.path.join(out, "meta_state.pdopt"))
    logger.info(f"saved distributed checkpoint to {out}")
    return out


def load(ckpt_dir: str, model: torch.nn.Module, optimizer=None) -> dict:
    path = os.path.join(ckpt_dir, _subdir())
    if not os.path.isdir(path):
        path = ckpt_dir
    sd = torch.load(os.path.join(path, "model.pdparams"),
                    map_location="cpu", weights_only=False)
    missing, unexpected = model.load_state_dict(sd, strict=False)
    if missing or unexpected:
        logger.warning(f"dist load: missing={missing} unexpected={unexpected}")
    if optimizer is not None:
        opt_path = os.path.join(path, "model_state.pdopt")
        if os.path.exists(opt_path):
            optimizer.load_state_dict(torch.load(opt_path, map_location="cpu",
                                                 weights_only=False))
    meta_path = os.path.join(path, "meta_state.pdopt")
    meta = {}
    if os.path.exists(meta_path):
        meta = torch.load(meta_path, map_location="cpu", weights_only=False)
        if "cpu_rng_state" in meta:
            torch.set_rng_state(meta["cpu_rng_state"])
        if "cuda_rng_state" in meta and torch.cuda.is_available():
            torch.cuda.set_rng_state(meta["cuda_rng_state"])
    logger.info(f"loaded distributed checkpoint from {path}")
    return meta
