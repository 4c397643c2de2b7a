"""Dual-domain RNG management (reference: parallel_layers/random.py:100-127).

Two seed domains: one shared across TP ranks of a DP replica (so replicated
params/dropout agree) and one offset per TP rank (so sharded params differ).
On ROCm the device generator is torch.cuda's; on CPU torch's default.
"""

import contextlib

import torch

from . import parallel_state as ps

_MODEL_PARALLEL_RNG_TRACKER_NAME = "model-parallel-rng"

_TP_SEED_OFFSET = 2718  # reference uses tp_rank-dependent offsets


class RNGStatesTracker:
    """Fork/restore named RNG states (reference XLARNGStatesTracker)."""

    def __init__(self):
        self.states_ = {}

    def reset(self):
        self.states_ = {}

    def add(self, name: str, seed: int):
        if name in self.states_:
            raise RuntimeError(f"rng state {name} already exists")
        orig = self._get_state()
        self._manual_seed(seed)
        self.states_[name] = self._get_state()
        self._set_state(orig)

    def _on_gpu(self):
        return torch.cuda.is_available() and torch.cuda.is_initialized()

    def _get_state(self):
        if self._on_gpu():
            return torch.cuda.get_rng_state()
        return torch.get_rng_state()

    def _set_state(self, state):
        if self._on_gpu():
            torch.cuda.set_rng_state(state)
        else:
            torch.set_rng_state(state)

    def _manual_seed(self, seed):
        if self._on_gpu():
            torch.cuda.manual_seed(seed)
        else:
            torch.manual_seed(seed)

    @contextlib.contextmanager
    def fork(self, name: str = _MODEL_PARALLEL_RNG_TRACKER_NAME):
        if name not in self.states_:
            raise RuntimeError(f"rng state {name} not added")
        orig = self._get_state()
        self._set_state(self.states_[name])
        try:
            yield
        finally:
            self.states_[name] = self._get_state()
            self._set_state(orig)


_RNG_STATE_TRACKER = RNGStatesTracker()


def get_rng_state_tracker() -> RNGStatesTracker:
    return _RNG_STATE_TRACKER


# Reference name (random.py:70-88)
get_xla_rng_tracker = get_rng_state_tracker


def model_parallel_manual_seed(seed: int):
    """Seed both domains (reference model_parallel_xla_manual_seed,
    random.py:100-127): data-parallel domain = seed (same on all TP ranks),
    model-parallel domain = seed + offset + tp_rank (different per TP rank).
    """
    tp_rank = ps.get_tensor_model_parallel_rank() if ps.model_parallel_is_initialized() else 0
    mp_seed = seed + _TP_SEED_OFFSET + tp_rank
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed(seed)
    _RNG_STATE_TRACKER.reset()
    _RNG_STATE_TRACKER.add(_MODEL_PARALLEL_RNG_TRACKER_NAME, mp_seed)


model_parallel_xla_manual_seed = model_parallel_manual_seed


def set_random_seed(seed: int):
    """Seed python/torch RNGs identically on every rank (reference
    utils/random.py:8); the model-parallel dual-domain seeding is
    ``model_parallel_manual_seed``."""
    import random as _random

    _random.seed(seed)
    try:
        import numpy as _np

        _np.random.seed(seed)
    except ImportError:
        pass
    import torch as _torch

    _torch.manual_seed(seed)
    if _torch.cuda.is_available():
        _torch.cuda.manual_seed_all(seed)
