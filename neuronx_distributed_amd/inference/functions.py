"""Inference fundamental units (reference trace/functions.py:390,470,808).

On MI355X the reference's trace->HLO->compile->NEFF chain collapses to
"build the eager module (+ optional hipGraph capture per bucket)", so:

* ``trace(model_fn, kwargs)``  -> a TraceArtifact recording the module
  factory + example inputs (shape-bucket key),
* ``compile(artifact)``        -> the artifact itself (layout/WLO has no
  MI355X analogue; kept as a no-op hook),
* ``shard_checkpoint(ckpt, model_fn, tp)`` -> per-rank sharded
  safetensors files ``tp{r}_sharded_checkpoint.safetensors`` using each
  module's preshard_hook / partition_dim attributes (reference
  functions.py:808,884-890; trace/trace.py:646-787).
"""

import os
from dataclasses import dataclass
from typing import Any, Callable, Dict, Optional

import torch

from ..parallel import parallel_state as ps
from .parallel_context import NxDParallelState


@dataclass
class TraceArtifact:
    model_fn: Callable
    example_inputs: Dict[str, Any]
    tag: str
    tp_degree: int
    compiled: bool = False
    compiler_args: Optional[dict] = None


def trace(model_fn: Callable, example_inputs: Dict[str, Any], tag: str = "",
          tp_degree: int = 1) -> TraceArtifact:
    return TraceArtifact(model_fn=model_fn, example_inputs=example_inputs,
                         tag=tag, tp_degree=tp_degree)


def compile(artifact: TraceArtifact, compiler_args=None) -> TraceArtifact:
    """MI355X: nothing to AOT-compile — hipGraph capture happens at first
    execution on device (NxDModel).  Kept for API parity; records args."""
    artifact.compiled = True
    artifact.compiler_args = compiler_args
    return artifact


def compile_wlo(artifact, *a, **k):
    """Weight-layout optimization has no MI355X analogue (we own the
    layouts); no-op hook (reference functions.py:597)."""
    return artifact


def compile_layout_transformer(artifact, *a, **k):
    """No-op (reference functions.py:689)."""
    return artifact


def shard_checkpoint(checkpoint: Dict[str, torch.Tensor], model_fn: Callable,
                     tp_degree: int, serialize_path: Optional[str] = None,
                     model_kwargs: Optional[dict] = None):
    """Split a FULL checkpoint into per-TP-rank shards by instantiating the
    model under NxDParallelState(rank=r) and applying preshard_hooks +
    partition_dim slicing.  Returns the list of shard dicts; writes
    safetensors when ``serialize_path`` given."""
    from ..parallel.checkpointing import _reshard_full_state_dict

    model_kwargs = model_kwargs or {}
    shards = []
    for r in range(tp_degree):
        with NxDParallelState(world_size=tp_degree, rank=r,
                              tensor_model_parallel_size=tp_degree):
            model = model_fn(**model_kwargs)
            sd = {k: v.clone() if isinstance(v, torch.Tensor) else v
                  for k, v in checkpoint.items()}
            _reshard_full_state_dict(model, sd)
        shards.append(sd)
        if serialize_path is not None:
            os.makedirs(serialize_path, exist_ok=True)
            from safetensors.torch import save_file

            save_file(
                {k: v.contiguous() for k, v in sd.items()
                 if isinstance(v, torch.Tensor)},
                os.path.join(serialize_path,
                             f"tp{r}_sharded_checkpoint.safetensors"))
        del model
    return shards


def load_sharded_checkpoint(serialize_path: str, tp_rank: int):
    from safetensors.torch import load_file

    return load_file(os.path.join(
        serialize_path, f"tp{tp_rank}_sharded_checkpoint.safetensors"))
