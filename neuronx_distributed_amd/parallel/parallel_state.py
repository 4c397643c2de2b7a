"""Process-group topology ("mpu") for the MI355X-native framework.

Role parity with the reference's ``parallel_layers/parallel_state.py``
(reference: parallel_state.py:391 ``initialize_model_parallel``,
:41-65 group globals, :620-636 mesh construction) — re-designed for one
process per MI355X GPU over RCCL (``torch.distributed`` backend "nccl" on
ROCm) with a gloo CPU mode for tests.

Topology model: the world is reshaped as the mesh ``[PP, DP, CP, TP]``
(TP innermost so TP groups are contiguous ranks and stay intra-node over
xGMI), plus the expert-parallel view ``[PP, DP_exp, EP, TP]`` where
``DP_exp * EP == DP * CP``.  Groups are first computed as plain rank
lists (``List[List[int]]``) so the topology is unit-testable without any
process group (``mesh_only=True``), mirroring the reference's
``mesh_only`` mode (parallel_state.py:398,673-674).

On an 8-GPU MI355X node the xGMI fabric is fully connected (7 links per
GPU), so the reference's trn1/trn2-specific rank placements (LOGIC2
ascending-descending rings, TP4 interleave) have no analogue here; plain
row-major placement (the reference's LOGIC1) is the xGMI-native layout.
"""

import os
from dataclasses import dataclass
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from ..utils.logger import get_logger

logger = get_logger(__name__)

# ---------------------------------------------------------------------------
# Globals (one per kind of group, mirroring the reference's module globals)
# ---------------------------------------------------------------------------

_TENSOR_MODEL_PARALLEL_GROUP = None
_DATA_PARALLEL_GROUP = None
_PIPELINE_MODEL_PARALLEL_GROUP = None
_CONTEXT_MODEL_PARALLEL_GROUP = None
_EXPERT_MODEL_PARALLEL_GROUP = None
_EXPERT_DATA_PARALLEL_GROUP = None
_TOKEN_SHUFFLE_GROUP = None
_KV_SHARED_GROUP = None

_NEXT_RANK_GROUP = None
_PREV_RANK_GROUP = None

_MPU_WORLD_SIZE: Optional[int] = None
_MPU_RANK: Optional[int] = None

_GROUPS: Dict[str, "GroupInfo"] = {}

_AOT_MODE = False  # single-process SPMD tracing mode (inference shard_checkpoint)
_AOT_WORLD_SIZE = None
_AOT_RANK = None


class GroupInfo:
    """A process group plus the rank-list mesh that defines it.

    The mesh (``List[List[int]]``) is the source of truth — it doubles as
    the SPMD/AOT representation the same way the reference attaches
    ``_mesh`` to its groups (reference: mock_torchdist.py:46-53).
    """

    def __init__(self, name: str, mesh: List[List[int]], group=None):
        self.name = name
        self.mesh = mesh
        self.group = group  # torch.distributed ProcessGroup or None (mesh_only)

    @property
    def size(self) -> int:
        return len(self.mesh[0]) if self.mesh else 1

    def ranks_of(self, rank: int) -> List[int]:
        for ranks in self.mesh:
            if rank in ranks:
                return ranks
        raise ValueError(f"rank {rank} not in any {self.name} group: {self.mesh}")

    def rank_in_group(self, rank: int) -> int:
        return self.ranks_of(rank).index(rank)


@dataclass
class ParallelGroups:
    """Pure rank-list result of topology construction (``mesh_only`` mode)."""

    tp_groups: List[List[int]]
    dp_groups: List[List[int]]
    pp_groups: List[List[int]]
    cp_groups: List[List[int]]
    ep_model_groups: List[List[int]]
    ep_data_groups: List[List[int]]
    kv_groups: Optional[List[List[int]]] = None
    token_shuffle_groups: Optional[List[List[int]]] = None
    zero1_groups: Optional[List[List[int]]] = None


def _build_mesh(world_size, tp, pp, cp, ep):
    """Return the rank-list meshes for every group kind.

    Mesh order is ``[PP, DP, CP, TP]`` (TP contiguous/innermost, PP
    outermost), matching the reference's layout decision
    (parallel_state.py:417-431,620-636) which keeps TP intra-node.
    """
    if world_size % (tp * pp * cp) != 0:
        raise ValueError(
            f"world size {world_size} not divisible by tp({tp})*pp({pp})*cp({cp})"
        )
    dp = world_size // (tp * pp * cp)
    if (dp * cp) % ep != 0:
        raise ValueError(f"dp*cp ({dp*cp}) not divisible by ep ({ep})")
    dp_exp = dp * cp // ep

    mesh = torch.arange(world_size).reshape(pp, dp, cp, tp)

    tp_groups = mesh.reshape(-1, tp).tolist()
    # DP: fix (pp, cp, tp) vary dp
    dp_groups = mesh.permute(0, 2, 3, 1).reshape(-1, dp).tolist()
    # PP: fix (dp, cp, tp) vary pp
    pp_groups = mesh.permute(1, 2, 3, 0).reshape(-1, pp).tolist()
    # CP: fix (pp, dp, tp) vary cp
    cp_groups = mesh.permute(0, 1, 3, 2).reshape(-1, cp).tolist()

    # Expert view: [PP, DP_exp, EP, TP]; EP and DP_exp carve up DPxCP
    emesh = mesh.reshape(pp, dp_exp, ep, tp)
    ep_model_groups = emesh.permute(0, 1, 3, 2).reshape(-1, ep).tolist()
    ep_data_groups = emesh.permute(0, 2, 3, 1).reshape(-1, dp_exp).tolist()

    # ZeRO-1 sharding groups: merged DPxCP (reference parallel_state.py:1684-1706)
    zero1_groups = mesh.permute(0, 3, 1, 2).reshape(-1, dp * cp).tolist()

    pg = ParallelGroups(
        tp_groups=tp_groups,
        dp_groups=dp_groups,
        pp_groups=pp_groups,
        cp_groups=cp_groups,
        ep_model_groups=ep_model_groups,
        ep_data_groups=ep_data_groups,
        zero1_groups=zero1_groups,
    )
    return pg


def _new_group(mesh: List[List[int]], name: str, backend=None) -> GroupInfo:
    """Create the torch.distributed groups for a mesh; return GroupInfo
    holding the group this rank belongs to."""
    rank = dist.get_rank()
    my_group = None
    for ranks in mesh:
        grp = dist.new_group(ranks, backend=backend)
        if rank in ranks:
            my_group = grp
    return GroupInfo(name, mesh, my_group)


def initialize_model_parallel(
    tensor_model_parallel_size: int = 1,
    pipeline_model_parallel_size: int = 1,
    context_parallel_size: int = 1,
    expert_model_parallel_size: int = 1,
    kv_size_multiplier: int = 1,
    token_shuffle_group_size: int = 1,
    mesh_only: bool = False,
    skip_collective_init: bool = False,
):
    """Build every parallel group (reference: parallel_state.py:391).

    With ``mesh_only=True`` no process groups are created and the pure
    rank-list :class:`ParallelGroups` is returned (unit-test mode,
    reference parallel_state.py:398).
    """
    global _TENSOR_MODEL_PARALLEL_GROUP, _DATA_PARALLEL_GROUP
    global _PIPELINE_MODEL_PARALLEL_GROUP, _CONTEXT_MODEL_PARALLEL_GROUP
    global _EXPERT_MODEL_PARALLEL_GROUP, _EXPERT_DATA_PARALLEL_GROUP
    global _TOKEN_SHUFFLE_GROUP, _KV_SHARED_GROUP
    global _NEXT_RANK_GROUP, _PREV_RANK_GROUP
    global _MPU_WORLD_SIZE, _MPU_RANK, _GROUPS

    tp = tensor_model_parallel_size
    pp = pipeline_model_parallel_size
    cp = context_parallel_size
    ep = expert_model_parallel_size

    if mesh_only:
        world_size = _MPU_WORLD_SIZE or (
            dist.get_world_size() if dist.is_initialized() else tp * pp * cp
        )
        return _build_mesh(world_size, tp, pp, cp, ep)

    if _AOT_MODE:
        world_size = _AOT_WORLD_SIZE
        groups = _build_mesh(world_size, tp, pp, cp, ep)
        _install_aot_groups(groups, tp, pp, cp, ep, kv_size_multiplier,
                            token_shuffle_group_size)
        return groups

    if not dist.is_initialized():
        raise RuntimeError(
            "torch.distributed must be initialized before initialize_model_parallel"
        )
    world_size = dist.get_world_size()
    groups = _build_mesh(world_size, tp, pp, cp, ep)

    if model_parallel_is_initialized():
        raise RuntimeError("model parallel already initialized")

    if not skip_collective_init:
        # Collective bootstrap: one tiny all-reduce warms RCCL communicator
        # creation before any real traffic (reference: parallel_state.py:647-657
        # warms a dummy all-reduce NEFF; here it forces RCCL rendezvous).
        t = torch.ones(1)
        if torch.cuda.is_available() and dist.get_backend() == "nccl":
            t = t.cuda()
        dist.all_reduce(t)

    _TENSOR_MODEL_PARALLEL_GROUP = _new_group(groups.tp_groups, "tp")
    _DATA_PARALLEL_GROUP = _new_group(groups.dp_groups, "dp")
    if cp > 1:
        zero1_info = _new_group(groups.zero1_groups, "zero1")
    else:
        zero1_info = GroupInfo("zero1", groups.zero1_groups,
                               _DATA_PARALLEL_GROUP.group)
    _PIPELINE_MODEL_PARALLEL_GROUP = _new_group(groups.pp_groups, "pp")
    _CONTEXT_MODEL_PARALLEL_GROUP = _new_group(groups.cp_groups, "cp")
    _EXPERT_MODEL_PARALLEL_GROUP = _new_group(groups.ep_model_groups, "ep")
    _EXPERT_DATA_PARALLEL_GROUP = _new_group(groups.ep_data_groups, "edp")

    if kv_size_multiplier > 1:
        kv_mesh = _build_kv_shared_mesh(groups.tp_groups, kv_size_multiplier)
        _KV_SHARED_GROUP = _new_group(kv_mesh, "kv")
        groups.kv_groups = kv_mesh
    if token_shuffle_group_size > 1:
        ts_mesh = _build_token_shuffle_mesh(groups.dp_groups, token_shuffle_group_size)
        _TOKEN_SHUFFLE_GROUP = _new_group(ts_mesh, "token_shuffle")
        groups.token_shuffle_groups = ts_mesh

    # PP neighbour pair groups (reference parallel_state.py:749-782): with
    # real RCCL/gloo P2P we use dist.send/recv directly, but batched
    # isend/irecv still wants a group handle per pair for ordering; we keep
    # the plain PP group and do P2P on global ranks.
    _MPU_WORLD_SIZE = world_size
    _MPU_RANK = dist.get_rank()

    _GROUPS = {
        "tp": _TENSOR_MODEL_PARALLEL_GROUP,
        "dp": _DATA_PARALLEL_GROUP,
        "pp": _PIPELINE_MODEL_PARALLEL_GROUP,
        "cp": _CONTEXT_MODEL_PARALLEL_GROUP,
        "ep": _EXPERT_MODEL_PARALLEL_GROUP,
        "edp": _EXPERT_DATA_PARALLEL_GROUP,
        "zero1": zero1_info,
    }
    if _KV_SHARED_GROUP is not None:
        _GROUPS["kv"] = _KV_SHARED_GROUP
    if _TOKEN_SHUFFLE_GROUP is not None:
        _GROUPS["token_shuffle"] = _TOKEN_SHUFFLE_GROUP

    logger.info(
        "initialized model parallel: tp=%d pp=%d cp=%d ep=%d dp=%d world=%d",
        tp, pp, cp, ep, world_size // (tp * pp * cp), world_size,
    )
    return groups


def _install_aot_groups(groups, tp, pp, cp, ep, kv_size_multiplier,
                        token_shuffle_group_size):
    """AOT (single-process SPMD tracing) mode: install GroupInfo objects with
    ``group=None``; collectives are mocked at the comm layer."""
    global _TENSOR_MODEL_PARALLEL_GROUP, _DATA_PARALLEL_GROUP
    global _PIPELINE_MODEL_PARALLEL_GROUP, _CONTEXT_MODEL_PARALLEL_GROUP
    global _EXPERT_MODEL_PARALLEL_GROUP, _EXPERT_DATA_PARALLEL_GROUP
    global _KV_SHARED_GROUP, _TOKEN_SHUFFLE_GROUP, _GROUPS, _MPU_WORLD_SIZE, _MPU_RANK
    _TENSOR_MODEL_PARALLEL_GROUP = GroupInfo("tp", groups.tp_groups)
    _DATA_PARALLEL_GROUP = GroupInfo("dp", groups.dp_groups)
    _PIPELINE_MODEL_PARALLEL_GROUP = GroupInfo("pp", groups.pp_groups)
    _CONTEXT_MODEL_PARALLEL_GROUP = GroupInfo("cp", groups.cp_groups)
    _EXPERT_MODEL_PARALLEL_GROUP = GroupInfo("ep", groups.ep_model_groups)
    _EXPERT_DATA_PARALLEL_GROUP = GroupInfo("edp", groups.ep_data_groups)
    if kv_size_multiplier > 1:
        groups.kv_groups = _build_kv_shared_mesh(groups.tp_groups, kv_size_multiplier)
        _KV_SHARED_GROUP = GroupInfo("kv", groups.kv_groups)
    _MPU_WORLD_SIZE = _AOT_WORLD_SIZE
    _MPU_RANK = _AOT_RANK
    _GROUPS = {
        "tp": _TENSOR_MODEL_PARALLEL_GROUP,
        "dp": _DATA_PARALLEL_GROUP,
        "pp": _PIPELINE_MODEL_PARALLEL_GROUP,
        "cp": _CONTEXT_MODEL_PARALLEL_GROUP,
        "ep": _EXPERT_MODEL_PARALLEL_GROUP,
        "edp": _EXPERT_DATA_PARALLEL_GROUP,
    }


def _build_kv_shared_mesh(tp_groups, kv_size_multiplier):
    """Ranks inside one TP group that hold replicas of the same KV head.

    Adjacent replication layout (reference qkv_linear.py:80-88; the trn1
    interleave is hardware-specific and not carried over): replicated slot
    ``s`` holds original head ``s // kv_size_multiplier``, so the replicas
    of one head sit on ``kv_size_multiplier`` CONTIGUOUS ranks of the TP
    group.
    """
    kv_mesh = []
    for ranks in tp_groups:
        for i in range(0, len(ranks), kv_size_multiplier):
            kv_mesh.append(ranks[i : i + kv_size_multiplier])
    return kv_mesh


def _build_token_shuffle_mesh(dp_groups, group_size):
    mesh = []
    for ranks in dp_groups:
        for i in range(0, len(ranks), group_size):
            mesh.append(ranks[i : i + group_size])
    return mesh


def model_parallel_is_initialized() -> bool:
    return _TENSOR_MODEL_PARALLEL_GROUP is not None


def destroy_model_parallel():
    global _TENSOR_MODEL_PARALLEL_GROUP, _DATA_PARALLEL_GROUP
    global _PIPELINE_MODEL_PARALLEL_GROUP, _CONTEXT_MODEL_PARALLEL_GROUP
    global _EXPERT_MODEL_PARALLEL_GROUP, _EXPERT_DATA_PARALLEL_GROUP
    global _TOKEN_SHUFFLE_GROUP, _KV_SHARED_GROUP, _GROUPS
    global _MPU_WORLD_SIZE, _MPU_RANK
    _TENSOR_MODEL_PARALLEL_GROUP = None
    _DATA_PARALLEL_GROUP = None
    _PIPELINE_MODEL_PARALLEL_GROUP = None
    _CONTEXT_MODEL_PARALLEL_GROUP = None
    _EXPERT_MODEL_PARALLEL_GROUP = None
    _EXPERT_DATA_PARALLEL_GROUP = None
    _TOKEN_SHUFFLE_GROUP = None
    _KV_SHARED_GROUP = None
    _GROUPS = {}
    _MPU_WORLD_SIZE = None
    _MPU_RANK = None


# ---------------------------------------------------------------------------
# Accessors
# ---------------------------------------------------------------------------

def _info(name: str) -> GroupInfo:
    if name not in _GROUPS:
        raise RuntimeError(
            f"model parallel group '{name}' not initialized; call "
            "initialize_model_parallel first"
        )
    return _GROUPS[name]


def _cur_rank() -> int:
    if _AOT_MODE:
        return _AOT_RANK
    return dist.get_rank()


def get_group_info(name: str) -> GroupInfo:
    return _info(name)


def get_tensor_model_parallel_group(as_list: bool = False):
    g = _info("tp")
    return g.mesh if as_list else g.group


def get_tensor_model_parallel_size() -> int:
    return _info("tp").size


# Megatron-compatible alias
get_tensor_model_parallel_world_size = get_tensor_model_parallel_size


def get_tensor_model_parallel_rank() -> int:
    return _info("tp").rank_in_group(_cur_rank())


def get_tensor_model_parallel_src_rank() -> int:
    return _info("tp").ranks_of(_cur_rank())[0]


def get_data_parallel_group(as_list: bool = False):
    g = _info("dp")
    return g.mesh if as_list else g.group


def get_data_parallel_size() -> int:
    return _info("dp").size


get_data_parallel_world_size = get_data_parallel_size


def get_data_parallel_rank() -> int:
    return _info("dp").rank_in_group(_cur_rank())


def get_pipeline_model_parallel_group(as_list: bool = False):
    g = _info("pp")
    return g.mesh if as_list else g.group


def get_pipeline_model_parallel_size() -> int:
    return _info("pp").size


get_pipeline_model_parallel_world_size = get_pipeline_model_parallel_size


def get_pipeline_model_parallel_rank() -> int:
    return _info("pp").rank_in_group(_cur_rank())


def get_pipeline_model_parallel_sharding_ranks() -> List[int]:
    return _info("pp").ranks_of(_cur_rank())


def get_pipeline_model_parallel_next_rank() -> int:
    ranks = _info("pp").ranks_of(_cur_rank())
    idx = ranks.index(_cur_rank())
    return ranks[(idx + 1) % len(ranks)]


def get_pipeline_model_parallel_prev_rank() -> int:
    ranks = _info("pp").ranks_of(_cur_rank())
    idx = ranks.index(_cur_rank())
    return ranks[(idx - 1) % len(ranks)]


def is_pipeline_first_stage() -> bool:
    return get_pipeline_model_parallel_rank() == 0


def is_pipeline_last_stage() -> bool:
    return get_pipeline_model_parallel_rank() == get_pipeline_model_parallel_size() - 1


def get_context_model_parallel_group(as_list: bool = False):
    g = _info("cp")
    return g.mesh if as_list else g.group


def get_context_model_parallel_size() -> int:
    return _info("cp").size


def get_context_model_parallel_rank() -> int:
    return _info("cp").rank_in_group(_cur_rank())


def get_expert_model_parallel_group(as_list: bool = False):
    g = _info("ep")
    return g.mesh if as_list else g.group


def get_expert_model_parallel_size() -> int:
    return _info("ep").size


def get_expert_model_parallel_rank() -> int:
    return _info("ep").rank_in_group(_cur_rank())


def get_expert_data_parallel_group(as_list: bool = False):
    g = _info("edp")
    return g.mesh if as_list else g.group


def get_expert_data_parallel_size() -> int:
    return _info("edp").size


def get_expert_data_parallel_rank() -> int:
    return _info("edp").rank_in_group(_cur_rank())


def get_kv_shared_group(as_list: bool = False):
    g = _info("kv")
    return g.mesh if as_list else g.group


def get_kv_shared_group_size() -> int:
    return _info("kv").size


def initialize_speculative_draft_group(draft_tp_size: int):
    """Spec-draft groups (reference parallel_state.py:1533): sub-groups of
    ``draft_tp_size`` ranks inside each TP group — a draft model usually
    runs at a smaller TP degree on a subset of the target's ranks.
    Callable any time after initialize_model_parallel (collective: every
    rank must call with the same size)."""
    global _GROUPS
    tp_mesh = _info("tp").mesh
    assert len(tp_mesh[0]) % draft_tp_size == 0, (
        f"draft tp {draft_tp_size} must divide tp {len(tp_mesh[0])}")
    mesh = []
    for row in tp_mesh:
        for i in range(0, len(row), draft_tp_size):
            mesh.append(row[i:i + draft_tp_size])
    _GROUPS["spec_draft"] = _new_group(mesh, "spec_draft")
    return _GROUPS["spec_draft"]


def get_speculative_draft_group(as_list: bool = False):
    g = _info("spec_draft")
    return g.mesh if as_list else g.group


def get_token_shuffle_group(as_list: bool = False):
    g = _info("token_shuffle")
    return g.mesh if as_list else g.group


def get_token_shuffle_group_size() -> int:
    return _info("token_shuffle").size


def get_zero1_sharding_group(as_list: bool = False):
    """Merged DPxCP group that ZeRO-1 shards over (reference
    parallel_state.py:1684-1706)."""
    g = _info("zero1")
    return g.mesh if as_list else g.group


def get_world_group():
    return None  # default group


# ---------------------------------------------------------------------------
# AOT / SPMD-trace mode (reference: parallel_state.py:101-102,1593-1602)
# ---------------------------------------------------------------------------

def enter_aot_mode(world_size: int, rank: int = 0):
    global _AOT_MODE, _AOT_WORLD_SIZE, _AOT_RANK
    _AOT_MODE = True
    _AOT_WORLD_SIZE = world_size
    _AOT_RANK = rank


def exit_aot_mode():
    global _AOT_MODE, _AOT_WORLD_SIZE, _AOT_RANK
    _AOT_MODE = False
    _AOT_WORLD_SIZE = None
    _AOT_RANK = None


def is_aot_mode() -> bool:
    return _AOT_MODE


# ---------------------------------------------------------------------------
# Rank-tagged logging (reference: parallel_state.py:1648-1667)
# ---------------------------------------------------------------------------

def rmsg(msg: str) -> str:
    """Prefix a message with this rank's position in every mesh."""
    if not model_parallel_is_initialized():
        r = dist.get_rank() if dist.is_initialized() else 0
        return f"[rank_{r}] {msg}"
    r = _cur_rank()
    return (
        f"[rank_{r}_pp{get_pipeline_model_parallel_rank()}"
        f"_tp{get_tensor_model_parallel_rank()}"
        f"_dp{get_data_parallel_rank()}"
        f"_cp{get_context_model_parallel_rank()}] {msg}"
    )
