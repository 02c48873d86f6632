"""Distributed environment bootstrap + seed discipline.

Reference surface: ppfleetx/distributed/apis/env.py
  init_dist_env (:121-151), get_hcg (:101-108), set_seed (:34-98),
  get_data_world_size (:158-166).

MI355X-native: one process per GPU, torch.distributed with the "nccl"
backend (= RCCL over xGMI on ROCm); "gloo" on CPU-only hosts so the
multi-process logic is testable without GPUs.
"""

from __future__ import annotations

import datetime
import os
import random
from typing import Optional

import numpy as np
import torch
import torch.distributed as dist

from paddlefleetx_amd.parallel.topology import HybridTopology
from paddlefleetx_amd.utils.log import logger

_HCG: Optional[HybridTopology] = None
_GLOBAL_SEED: Optional[int] = None
_LOCAL_SEED: Optional[int] = None


def get_hcg() -> HybridTopology:
    global _HCG
    if _HCG is None:
        _HCG = HybridTopology()  # degenerate single-rank topology
    return _HCG


def set_hcg(hcg: HybridTopology) -> None:
    global _HCG
    _HCG = hcg


def world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def init_process_group(backend: Optional[str] = None, timeout_sec: int = 1800):
    if dist.is_initialized():
        return
    if "RANK" not in os.environ:
        return  # single-process run
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if backend == "nccl":
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        torch.cuda.set_device(local_rank)
    dist.init_process_group(backend=backend,
                            timeout=datetime.timedelta(seconds=timeout_sec))


def init_dist_env(config) -> HybridTopology:
    """Build process groups from config.Distributed (reference env.py:121-151)."""
    init_process_group()
    d = config.get("Distributed", {})
    moe = bool(config.get("Model", {}).get("moe_configs", None))
    hcg = HybridTopology(
        dp=int(d.get("dp_degree", 1)),
        mp=int(d.get("mp_degree", 1)),
        pp=int(d.get("pp_degree", 1)),
        sharding=int(d.get("sharding", {}).get("sharding_degree", 1)),
        cp=int(d.get("cp_degree", 1) or 1),
        moe_expert_parallel=moe,
    )
    set_hcg(hcg)
    if hcg.global_rank == 0:
        logger.info(f"initialized {hcg}")
    seed = int(config.get("Global", {}).get("seed", 1024))
    set_seed(seed)
    return hcg


def set_seed(seed: int) -> None:
    """Deterministic seed scheme (reference env.py:34-98).

    global_seed: equal across mp ranks of one (pp, dp, sharding) replica so
      non-parallel randomness (data order, init of replicated weights) matches.
    local_seed: distinct per mp rank, used by the TP RNG tracker so dropout
      inside tensor-parallel regions differs per shard.
    """
    global _GLOBAL_SEED, _LOCAL_SEED
    hcg = get_hcg()
    global_seed = (seed + 100003 * hcg.pp_rank + 911 * hcg.dp_rank
                   + 137 * hcg.sharding_rank)
    local_seed = global_seed + 2717 * (hcg.mp_rank + 1)
    _GLOBAL_SEED, _LOCAL_SEED = global_seed, local_seed

    random.seed(global_seed)
    np.random.seed(global_seed % (2 ** 31))
    torch.manual_seed(global_seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed(global_seed)

    from paddlefleetx_amd.parallel.rng import get_rng_tracker
    tracker = get_rng_tracker()
    tracker.reset()
    tracker.add("local_seed", local_seed)
    tracker.add("global_seed", global_seed + 1)


def get_global_seed() -> Optional[int]:
    return _GLOBAL_SEED


def get_local_seed() -> Optional[int]:
    return _LOCAL_SEED


def get_data_world_size() -> int:
    return get_hcg().get_data_world_size()


def get_data_world_rank() -> int:
    return get_hcg().get_data_world_rank()
