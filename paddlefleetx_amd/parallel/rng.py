"""Model-parallel RNG state tracker.

Dropout inside tensor-parallel regions must use a per-mp-rank seed while
replicated regions share the global seed (reference env.py:34-98 +
hybrid_model.py:328/651/664 `get_rng_state_tracker`). Implemented by
checkpointing/restoring torch (and torch.cuda) RNG states around a `fork`.
"""

from __future__ import annotations

import contextlib
from typing import Dict

import torch

__all__ = ["get_rng_tracker", "model_parallel_rng"]


class RNGStateTracker:
    def __init__(self):
        self._states: Dict[str, tuple] = {}

    def reset(self):
        self._states.clear()

    def add(self, name: str, seed: int):
        cpu_state = torch.get_rng_state()
        cuda_state = torch.cuda.get_rng_state() if torch.cuda.is_available() else None
        torch.manual_seed(seed)
        if torch.cuda.is_available():
            torch.cuda.manual_seed(seed)
        self._states[name] = (torch.get_rng_state(),
                              torch.cuda.get_rng_state() if torch.cuda.is_available() else None)
        torch.set_rng_state(cpu_state)
        if cuda_state is not None:
            torch.cuda.set_rng_state(cuda_state)

    @contextlib.contextmanager
    def fork(self, name: str = "local_seed"):
        if name not in self._states:
            # tracker unseeded (single-card tests): plain passthrough
            yield
            return
        orig_cpu = torch.get_rng_state()
        orig_cuda = torch.cuda.get_rng_state() if torch.cuda.is_available() else None
        s_cpu, s_cuda = self._states[name]
        torch.set_rng_state(s_cpu)
        if s_cuda is not None:
            torch.cuda.set_rng_state(s_cuda)
        try:
            yield
        finally:
            self._states[name] = (torch.get_rng_state(),
                                  torch.cuda.get_rng_state() if torch.cuda.is_available() else None)
            torch.set_rng_state(orig_cpu)
            if orig_cuda is not None:
                torch.cuda.set_rng_state(orig_cuda)

    def state_dict(self):
        return dict(self._states)

    def load_state_dict(self, sd):
        self._states = dict(sd)


_TRACKER = RNGStateTracker()


def get_rng_tracker() -> RNGStateTracker:
    return _TRACKER


@contextlib.contextmanager
def model_parallel_rng():
    with _TRACKER.fork("local_seed"):
        yield
