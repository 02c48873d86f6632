"""Profiler wrapper: torch.profiler with the reference's YAML surface.

Reference: Profiler config section (pretrain_gpt_base.yaml:94-99) ->
paddle.profiler started in engine init (eager_engine.py:250-272),
`.step()` per iteration (:419-420), end-of-run summary (:866-925).

MI355X: torch.profiler on ROCm records HIP kernel events via roctracer;
chrome traces land in `profiler_log/` and open in perfetto. For per-kernel
hardware counters use rocprofv3 externally (profiles/README.md).
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from paddlefleetx_amd.utils.log import logger


class ProfilerGuard:
    """Engine-facing wrapper; inert unless Profiler.enable is True."""

    def __init__(self, config: Optional[dict]):
        cfg = dict(config or {})
        self.enabled = bool(cfg.get("enable", False))
        self.prof = None
        if not self.enabled:
            return
        sched = cfg.get("scheduler", [1, 5]) or [1, 5]
        wait = int(sched[0])
        active = int(sched[1]) - wait if len(sched) > 1 else 4
        self.log_dir = cfg.get("profiler_log", "profiler_log")
        self.record_shapes = bool(cfg.get("record_shapes", True))
        self.profile_memory = bool(cfg.get("profile_memory", True))
        self.detailed = bool(cfg.get("detailed", False))
        activities = [torch.profiler.ProfilerActivity.CPU]
        if torch.cuda.is_available():
            activities.append(torch.profiler.ProfilerActivity.CUDA)
        self.prof = torch.profiler.profile(
            activities=activities,
            schedule=torch.profiler.schedule(wait=wait, warmup=1,
                                             active=max(1, active)),
            on_trace_ready=torch.profiler.tensorboard_trace_handler(
                self.log_dir),
            record_shapes=self.record_shapes,
            profile_memory=self.profile_memory,
            with_stack=self.detailed)
        self.prof.start()
        logger.info(f"profiler enabled -> chrome traces in {self.log_dir}/")

    def step(self):
        if self.prof is not None:
            self.prof.step()

    def stop_and_summary(self):
        if self.prof is None:
            return
        try:
            self.prof.stop()
        except Exception:
            pass
        try:
            key = "cuda_time_total" if torch.cuda.is_available() \
                else "cpu_time_total"
            table = self.prof.key_averages().table(sort_by=key, row_limit=20)
            logger.info("profiler summary (top 20 by device time):\n" + table)
        except Exception as e:  # pragma: no cover
            logger.warning(f"profiler summary failed: {e}")
        self.prof = None
