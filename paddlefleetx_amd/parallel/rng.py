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


def checkpoint_rng_context():
    """`context_fn` for torch.utils.checkpoint(use_reentrant=False):
    checkpoint preserves only the GLOBAL torch/cuda RNG, so dropout (and
    the flash-dropout seed draw) inside a `model_parallel_rng()` fork
    would consume a FRESH tracker stream on the recompute re-forward —
    different masks forward vs backward, silently wrong gradients. The
    forward context snapshots the tracker; the recompute context rewinds
    to that snapshot for the replay and then puts the advanced states
    back, so the net stream position matches a no-recompute run."""
    snapshot: Dict[str, tuple] = {}

    def _clone(states):
        return {k: (c.clone(), None if g is None else g.clone())
                for k, (c, g) in states.items()}

    class _Forward(contextlib.AbstractContextManager):
        def __enter__(self):
            snapshot.clear()
            snapshot.update(_clone(_TRACKER._states))
            return self

        def __exit__(self, *exc):
            return False

    class _Recompute(contextlib.AbstractContextManager):
        def __enter__(self):
            self._advanced = _TRACKER._states
            _TRACKER._states = _clone(snapshot)
            return self

        def __exit__(self, *exc):
            _TRACKER._states = self._advanced
            return False

    return _Forward(), _Recompute()
