"""Low-level sharded checkpoint save/load.

Parity with reference ``parallel_layers/checkpointing.py`` (271 LoC):
per-rank files ``tp_rank_xx_pp_rank_xxx`` (:70-143), staggered serial load
(:218-244), ``preshard_hook`` invocation + full-checkpoint resharding
(:35-67).  The directory layout and the ``partition_dim`` attributes are
part of the public sharded-checkpoint contract.
"""

import os
from typing import Optional

import torch

from . import comm
from . import parallel_state as ps
from .utils import create_local_weight
from ..utils.logger import get_logger

logger = get_logger(__name__)


def _chkpt_path(path: str, tag: str) -> str:
    tp = ps.get_tensor_model_parallel_rank()
    pp = ps.get_pipeline_model_parallel_rank()
    ep = ps.get_expert_model_parallel_rank() if "ep" in ps._GROUPS else 0
    name = f"tp_rank_{tp:02d}_pp_rank_{pp:02d}"
    if ps.get_expert_model_parallel_size() > 1:
        name += f"_ep_rank_{ep:02d}"
    return os.path.join(path, tag, f"{name}.pt")


def save(state_dict: dict, path: str, tag: str = "model",
         save_serially: bool = False, down_cast_bf16: bool = False) -> None:
    """Save this rank's shard (only DP/CP rank 0 of each model-parallel
    replica writes; reference checkpointing.py:70-143)."""
    should_write = (ps.get_data_parallel_rank() == 0
                    and ps.get_context_model_parallel_size() == 1
                    or ps.get_data_parallel_rank() == 0
                    and ps.get_context_model_parallel_rank() == 0)
    fname = _chkpt_path(path, tag)
    os.makedirs(os.path.dirname(fname), exist_ok=True)
    if down_cast_bf16:
        state_dict = {
            k: (v.bfloat16() if isinstance(v, torch.Tensor) and v.is_floating_point()
                else v) for k, v in state_dict.items()
        }
    if should_write:
        cpu_state = {
            k: (v.cpu() if isinstance(v, torch.Tensor) else v)
            for k, v in state_dict.items()
        }
        torch.save(cpu_state, fname)
    comm.barrier()


def load(path: str, tag: str = "model", model: Optional[torch.nn.Module] = None,
         model_or_optimizer=None, sharded: bool = True, strict: bool = True,
         load_serially: bool = True):
    """Load this rank's shard; with ``sharded=False`` a FULL checkpoint is
    resharded on the fly via each module's ``preshard_hook``/partition_dim
    attributes (reference checkpointing.py:35-67,218-271)."""
    if model_or_optimizer is None:
        model_or_optimizer = model
    if sharded:
        fname = _chkpt_path(path, tag)
    else:
        fname = path
    if load_serially:
        # stagger loads over TP ranks to bound host RAM (reference :218-244)
        for r in range(ps.get_tensor_model_parallel_size()):
            if r == ps.get_tensor_model_parallel_rank():
                state_dict = torch.load(fname, map_location="cpu",
                                        weights_only=False)
            comm.barrier()
    else:
        state_dict = torch.load(fname, map_location="cpu", weights_only=False)

    if not sharded and model_or_optimizer is not None:
        _reshard_full_state_dict(model_or_optimizer, state_dict)

    if model_or_optimizer is not None:
        if isinstance(model_or_optimizer, torch.nn.Module):
            model_or_optimizer.load_state_dict(state_dict, strict=strict)
        else:
            model_or_optimizer.load_state_dict(state_dict)
        return model_or_optimizer
    return state_dict


def _reshard_full_state_dict(model: torch.nn.Module, state_dict: dict) -> None:
    """Invoke preshard_hooks, then slice any remaining full-size tensors by
    their target param's partition_dim (reference create_local_weight path)."""
    for name, module in model.named_modules():
        hook = getattr(module, "preshard_hook", None)
        if hook is not None and hasattr(module, "weight"):
            key = f"{name}.weight" if name else "weight"
            if key in state_dict:
                hook(state_dict, key)
    params = dict(model.named_parameters())
    for key, value in list(state_dict.items()):
        p = params.get(key)
        if p is None or not isinstance(value, torch.Tensor):
            continue
        pdim = getattr(p, "partition_dim", -1)
        if pdim >= 0 and getattr(p, "tensor_model_parallel", False) \
                and value.shape != p.shape:
            state_dict[key] = create_local_weight(
                value, pdim, p.shape[pdim], getattr(p, "partition_stride", 1))
