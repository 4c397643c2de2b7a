"""Distributed top-k over a TP-sharded dim (reference operators/topk.py:31):
local top-k per shard -> all-gather candidates -> global top-k with index
correction (cheaper than gathering the whole vocab)."""

import torch

from ..parallel import comm, parallel_state as ps


def topk(tensor: torch.Tensor, k: int, dim: int = -1, gather_dim: int = -1):
    """tensor (..., V/tp) sharded on gather_dim; returns (values, global
    indices)."""
    tp = ps.get_group_info("tp")
    world = tp.size
    if world == 1:
        return torch.topk(tensor, k, dim=dim)
    if dim != gather_dim:
        raise NotImplementedError("topk: dim must equal gather_dim")
    local_size = tensor.shape[dim]
    lv, li = torch.topk(tensor, min(k, local_size), dim=dim)
    rank = comm.group_rank(tp)
    li = li + rank * local_size
    all_v = comm.all_gather(lv, dim=dim, group=tp)
    all_i = comm.all_gather(li, dim=dim, group=tp)
    gv, gi = torch.topk(all_v, k, dim=dim)
    return gv, all_i.gather(dim, gi)
