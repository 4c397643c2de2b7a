"""Collective-communication wrapper layer.

Role parity with the reference's ``parallel_layers/comm.py`` (221 LoC,
XLA-vs-gloo dual dispatch, comm.py:124-220).  Here there is exactly ONE
device path — RCCL over xGMI via ``torch.distributed`` (backend "nccl"
IS RCCL on ROCm) — plus:

* a gloo CPU fallback for the ops gloo lacks (``reduce_scatter_tensor``,
  ``all_to_all_single``) so multi-process CPU tests of distributed
  semantics work (reference NXD_CPU_MODE, comm.py:32-121), and
* an AOT mode (single-process SPMD tracing) where collectives become
  identity/zero ops, mirroring the reference's mocked torch.distributed
  (trace/mock_torchdist.py:8-84).

Every function takes a :class:`GroupInfo` or raw ProcessGroup.
"""


import os

import torch
import torch.distributed as dist

from . import parallel_state as ps


def _unwrap(group):
    if isinstance(group, ps.GroupInfo):
        return group.group
    return group


def _backend_is_gloo(group) -> bool:
    if not dist.is_initialized():
        return False
    try:
        return dist.get_backend(group) == "gloo"
    except Exception:
        return False


def group_size(group) -> int:
    if isinstance(group, ps.GroupInfo):
        return group.size
    return dist.get_world_size(group=group)


def group_rank(group) -> int:
    if isinstance(group, ps.GroupInfo):
        return group.rank_in_group(dist.get_rank() if dist.is_initialized() else 0)
    return dist.get_rank(group=group)


def all_reduce(tensor: torch.Tensor, op=dist.ReduceOp.SUM, group=None,
               async_op: bool = False):
    """In-place all-reduce. Returns the async work handle if requested.

    NXDA_ONESHOT_AR_MAX_BYTES > 0 routes SUM all-reduces up to that size
    through a one-shot all-gather + local reduce: on point-to-point xGMI
    a ring all-reduce costs 2(N-1) latency steps on the per-link bound;
    one-shot is a single exchange and wins for small latency-bound
    payloads (SURVEY §7 hard-parts; default off, tune on 8 GPUs)."""
    if ps.is_aot_mode():
        return None
    if isinstance(group, ps.GroupInfo) and group.size == 1:
        return None
    g = _unwrap(group)
    limit = int(os.environ.get("NXDA_ONESHOT_AR_MAX_BYTES", "0"))
    if (limit > 0 and not async_op and op == dist.ReduceOp.SUM
            and tensor.numel() * tensor.element_size() <= limit):
        world = dist.get_world_size(group=g)
        flat = tensor.reshape(-1)
        gathered = torch.empty(world * flat.numel(), dtype=tensor.dtype,
                               device=tensor.device)
        dist.all_gather_into_tensor(gathered, flat.contiguous(), group=g)
        flat.copy_(gathered.view(world, -1).sum(dim=0))
        return None
    return dist.all_reduce(tensor, op=op, group=g, async_op=async_op)


def all_gather(tensor: torch.Tensor, dim: int = 0, group=None) -> torch.Tensor:
    """All-gather along ``dim`` (out-of-place).

    Dim-general via the transpose-to-0 trick (reference mappings.py:27-40):
    RCCL's all_gather_into_tensor concatenates on dim 0, so we move ``dim``
    to the front, gather, and move back.
    """
    if ps.is_aot_mode():
        return tensor
    g = _unwrap(group)
    world = dist.get_world_size(group=g)
    if world == 1:
        return tensor
    t = tensor if dim == 0 else tensor.movedim(dim, 0)
    t = t.contiguous()
    out = torch.empty((world,) + tuple(t.shape), dtype=t.dtype, device=t.device)
    if _backend_is_gloo(g):
        parts = [out[i] for i in range(world)]
        dist.all_gather(parts, t, group=g)
    else:
        dist.all_gather_into_tensor(out.view(-1), t.view(-1), group=g)
    out = out.reshape((world * t.shape[0],) + tuple(t.shape[1:]))
    if dim != 0:
        out = out.movedim(0, dim)
    return out.contiguous() if dim != 0 else out


def reduce_scatter(tensor: torch.Tensor, dim: int = 0, group=None,
                   op=dist.ReduceOp.SUM) -> torch.Tensor:
    """Reduce-scatter along ``dim`` (out-of-place).

    gloo has no reduce_scatter_tensor — emulate with all_reduce + slice
    (reference comm.py:32-121 emulates with reduce+scatter).
    """
    if ps.is_aot_mode():
        return tensor
    g = _unwrap(group)
    world = dist.get_world_size(group=g)
    if world == 1:
        return tensor
    t = tensor if dim == 0 else tensor.movedim(dim, 0)
    t = t.contiguous()
    assert t.shape[0] % world == 0, (
        f"reduce_scatter dim {dim} size {t.shape[0]} not divisible by {world}"
    )
    shard = t.shape[0] // world
    if _backend_is_gloo(g):
        dist.all_reduce(t, op=op, group=g)
        rank = dist.get_rank(group=g)
        out = t[rank * shard : (rank + 1) * shard].clone()
    else:
        out = torch.empty((shard,) + tuple(t.shape[1:]), dtype=t.dtype,
                          device=t.device)
        dist.reduce_scatter_tensor(out, t, op=op, group=g)
    if dim != 0:
        out = out.movedim(0, dim).contiguous()
    return out


def all_to_all(tensor: torch.Tensor, split_dim: int, concat_dim: int,
               group=None) -> torch.Tensor:
    """All-to-all: split on ``split_dim``, concatenate on ``concat_dim``
    (reference mappings.py:160-172)."""
    if ps.is_aot_mode():
        return tensor
    g = _unwrap(group)
    world = dist.get_world_size(group=g)
    if world == 1:
        return tensor
    assert tensor.shape[split_dim] % world == 0
    chunks = [c.contiguous() for c in tensor.chunk(world, dim=split_dim)]
    if _backend_is_gloo(g):
        # gloo has no alltoall: emulate with all_gather + chunk select
        rank = dist.get_rank(group=g)
        full = tensor.contiguous()
        gathered = [torch.empty_like(full) for _ in range(world)]
        dist.all_gather(gathered, full, group=g)
        outs = [gathered[r].chunk(world, dim=split_dim)[rank].contiguous()
                for r in range(world)]
    else:
        outs = [torch.empty_like(c) for c in chunks]
        dist.all_to_all(outs, chunks, group=g)
    return torch.cat(outs, dim=concat_dim)


def broadcast(tensor: torch.Tensor, src: int, group=None):
    if ps.is_aot_mode():
        return tensor
    g = _unwrap(group)
    dist.broadcast(tensor, src=src, group=g)
    return tensor


def barrier(group=None):
    if ps.is_aot_mode() or not dist.is_initialized():
        return
    dist.barrier(group=_unwrap(group))


def send(tensor: torch.Tensor, dst: int, group=None):
    dist.send(tensor, dst=dst, group=_unwrap(group))


def recv(tensor: torch.Tensor, src: int, group=None):
    dist.recv(tensor, src=src, group=_unwrap(group))
