"""ZeRO-1 optimizer state through torch.distributed.checkpoint (DCP).

Parity with the reference's ``optimizer/zero_dcp_utils.py`` (SURVEY.md
§2.4): there the zero1 shards are hand-wrapped as ShardedTensors with a
hand-driven planner because DCP's collectives could not run under XLA.
On MI355X the gloo/RCCL store works, so this is vanilla DCP over DTensors:
each flat master/momentum shard is declared as a 1-D DTensor sharded over
the zero1 group, and DCP's planner handles dedup, file layout and —
the point of the exercise — RESHARDING on load when the world size
changed between save and load.

The non-DCP path (per-rank files + offline merge in
``scripts/checkpoint_converter.py``) remains the default engine;
this module is the DCP alternative.
"""

import os
from typing import Dict

import torch
import torch.distributed as dist

from ..parallel import parallel_state as ps


def _zero1_mesh():
    from torch.distributed.device_mesh import init_device_mesh

    mesh_rows = ps.get_zero1_sharding_group(as_list=True)
    world = dist.get_world_size()
    if len(mesh_rows) != 1 or len(mesh_rows[0]) != world:
        raise NotImplementedError(
            "zero1 DCP save requires the zero1 sharding group to span the "
            "world (tp=pp=1); use the per-rank checkpoint engine + offline "
            "merge otherwise")
    device_type = "cuda" if torch.cuda.is_available() else "cpu"
    return init_device_mesh(device_type, (world,))


def _dtensor_state(optimizer) -> Dict[str, torch.Tensor]:
    """The zero1 shards as DTensors + replicated scalars."""
    from torch.distributed.tensor import DTensor, Shard

    mesh = _zero1_mesh()
    state: Dict[str, torch.Tensor] = {}
    for i, b in enumerate(optimizer.buckets):
        local = b.master.detach()
        state[f"bucket_{i}.master"] = DTensor.from_local(
            local, mesh, [Shard(0)])
        if hasattr(b, "fused_m"):
            state[f"bucket_{i}.m"] = DTensor.from_local(
                b.fused_m.detach(), mesh, [Shard(0)])
            state[f"bucket_{i}.v"] = DTensor.from_local(
                b.fused_v.detach(), mesh, [Shard(0)])
    state["step_count"] = torch.tensor(
        getattr(optimizer, "_step_count", 0))
    return state


def save_zero1_optimizer_dcp(optimizer, path: str) -> None:
    import torch.distributed.checkpoint as dcp

    os.makedirs(path, exist_ok=True)
    dcp.save(_dtensor_state(optimizer), checkpoint_id=path)


def load_zero1_optimizer_dcp(optimizer, path: str) -> None:
    """Loads (and reshards if the zero1 world changed) into the
    optimizer's live buckets, then re-broadcasts the bf16 params."""
    import torch.distributed.checkpoint as dcp

    state = _dtensor_state(optimizer)
    dcp.load(state, checkpoint_id=path)
    for i, b in enumerate(optimizer.buckets):
        b.master.data.copy_(state[f"bucket_{i}.master"].to_local())
        if hasattr(b, "fused_m"):
            b.fused_m.copy_(state[f"bucket_{i}.m"].to_local())
            b.fused_v.copy_(state[f"bucket_{i}.v"].to_local())
    optimizer._step_count = int(state["step_count"].item())
    optimizer._all_gather_params()
