"""Inference-model export.

Reference: ppfleetx/utils/export.py:85-150 (dy2static paddle.jit.save with
pruned input spec) + InferenceEngine's rank_{i}/ model layout
(core/engine/inference_engine.py:144-171).

MI355X-native: the deployable artifact is {rank_i/model.safetensors +
config.json}; the inference engine rebuilds the network from config and
loads the shard for its mp rank — no static-graph compiler, the HIP
kernels ARE the optimized path.
"""

from __future__ import annotations

import json
import os
from typing import Any, Dict

import torch

from paddlefleetx_amd.parallel.env import get_hcg
from paddlefleetx_amd.utils.log import logger


def export_inference_model(model: torch.nn.Module, model_cfg: Dict[str, Any],
                           out_dir: str, extra: Dict[str, Any] = None) -> str:
    """Write rank_{mp}/model.safetensors + config.json."""
    hcg = get_hcg()
    rank_dir = os.path.join(out_dir, f"rank_{hcg.get_model_parallel_rank()}")
    os.makedirs(rank_dir, exist_ok=True)
    sd = {k: v.detach().cpu() for k, v in model.state_dict().items()}
    try:
        from safetensors.torch import save_file
        # safetensors refuses shared storage (tied embeddings) - clone
        sd = {k: v.clone().contiguous() for k, v in sd.items()}
        save_file(sd, os.path.join(rank_dir, "model.safetensors"))
    except ImportError:
        torch.save(sd, os.path.join(rank_dir, "model.pt"))
    if hcg.get_model_parallel_rank() == 0:
        meta = {"model": model_cfg,
                "mp_degree": hcg.get_model_parallel_world_size()}
        meta.update(extra or {})
        with open(os.path.join(out_dir, "config.json"), "w") as f:
            json.dump(meta, f, indent=2, default=str)
    logger.info(f"exported inference model to {rank_dir}")
    return rank_dir


def load_inference_model(model: torch.nn.Module, model_dir: str) -> None:
    hcg = get_hcg()
    rank_dir = os.path.join(model_dir,
                            f"rank_{hcg.get_model_parallel_rank()}")
    st = os.path.join(rank_dir, "model.safetensors")
    if os.path.exists(st):
        from safetensors.torch import load_file
        sd = load_file(st)
    else:
        sd = torch.load(os.path.join(rank_dir, "model.pt"),
                        map_location="cpu", weights_only=False)
    missing, unexpected = model.load_state_dict(sd, strict=False)
    if missing or unexpected:
        logger.warning(f"inference load: missing={missing} "
                       f"unexpected={unexpected}")
