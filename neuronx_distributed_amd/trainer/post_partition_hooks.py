"""Post-partition hook registry (reference trainer/post_partition_hooks.py:5):
callables invoked on the model after the pipeline partition (phase 2) and
materialization — the place to re-tie weights, patch buffers, or register
custom state-dict transforms that need the FINAL module structure."""

from typing import Callable, List

_HOOKS: List[Callable] = []


def register_post_partition_hook(fn: Callable) -> Callable:
    """fn(model, nxd_config) -> None; returns fn (usable as decorator)."""
    _HOOKS.append(fn)
    return fn


def clear_post_partition_hooks() -> None:
    _HOOKS.clear()


def run_post_partition_hooks(model, nxd_config) -> None:
    for fn in list(_HOOKS):
        fn(model, nxd_config)
