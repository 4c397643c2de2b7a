"""Hybrid MoE sharding process groups (reference
modules/moe/moe_process_group.py:12-74).

Inference MoE layers may shard DIFFERENTLY for context encoding (CTE /
prefill: big token counts, usually TP-heavy) and token generation (TKG /
decode: latency-bound, often EP-heavy).  This module builds one (TP, EP)
group pair per phase from the mesh_only topology and hands the right one
out by ``prefill`` flag.

Group construction is collective (``dist.new_group`` over every mesh row)
exactly like ``parallel_state._new_group``.
"""

from typing import Optional

import torch.distributed as dist

from ..parallel import parallel_state as ps

_MOE_TKG_TP_GROUP: Optional[ps.GroupInfo] = None
_MOE_TKG_EP_GROUP: Optional[ps.GroupInfo] = None
_MOE_CTE_TP_GROUP: Optional[ps.GroupInfo] = None
_MOE_CTE_EP_GROUP: Optional[ps.GroupInfo] = None


def _build_pair(tp_degree: int, ep_degree: int, name: str):
    world = dist.get_world_size()
    mesh = ps._build_mesh(world, tp_degree, 1, 1, ep_degree)
    rank = dist.get_rank()

    def new_group(rows, suffix):
        mine = None
        for ranks in rows:
            g = dist.new_group(ranks)
            if rank in ranks:
                mine = g
        return ps.GroupInfo(f"{name}_{suffix}", rows, mine)

    return (new_group(mesh.tp_groups, "tp"),
            new_group(mesh.ep_model_groups, "ep"))


def init_tensor_expert_parallel_moe_process_groups(
        tkg_tp_degree: int, tkg_ep_degree: int,
        cte_tp_degree: int, cte_ep_degree: int) -> None:
    """Build both phase group pairs (idempotent)."""
    global _MOE_TKG_TP_GROUP, _MOE_TKG_EP_GROUP
    global _MOE_CTE_TP_GROUP, _MOE_CTE_EP_GROUP
    if _MOE_TKG_TP_GROUP is None and _MOE_TKG_EP_GROUP is None:
        _MOE_TKG_TP_GROUP, _MOE_TKG_EP_GROUP = _build_pair(
            tkg_tp_degree, tkg_ep_degree, "moe_tkg")
    if _MOE_CTE_TP_GROUP is None and _MOE_CTE_EP_GROUP is None:
        _MOE_CTE_TP_GROUP, _MOE_CTE_EP_GROUP = _build_pair(
            cte_tp_degree, cte_ep_degree, "moe_cte")


def get_moe_tp_ep_group(prefill: bool = True) -> ps.GroupInfo:
    g = _MOE_CTE_TP_GROUP if prefill else _MOE_TKG_TP_GROUP
    assert g is not None, (
        "MoE process groups not initialized — call "
        "init_tensor_expert_parallel_moe_process_groups first")
    return g


def get_moe_ep_group(prefill: bool = True) -> ps.GroupInfo:
    g = _MOE_CTE_EP_GROUP if prefill else _MOE_TKG_EP_GROUP
    assert g is not None, (
        "MoE process groups not initialized — call "
        "init_tensor_expert_parallel_moe_process_groups first")
    return g


def destroy_moe_model_parallel() -> None:
    global _MOE_TKG_TP_GROUP, _MOE_TKG_EP_GROUP
    global _MOE_CTE_TP_GROUP, _MOE_CTE_EP_GROUP
    _MOE_TKG_TP_GROUP = None
    _MOE_TKG_EP_GROUP = None
    _MOE_CTE_TP_GROUP = None
    _MOE_CTE_EP_GROUP = None
