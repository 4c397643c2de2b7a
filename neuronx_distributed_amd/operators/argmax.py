"""Distributed argmax over a TP-sharded dim (reference operators/argmax.py:55)."""

import torch

from ..parallel import comm, parallel_state as ps


def argmax(tensor: torch.Tensor, dim: int = -1, gather_dim: int = -1,
           keepdim: bool = False):
    tp = ps.get_group_info("tp")
    world = tp.size
    if world == 1:
        return torch.argmax(tensor, dim=dim, keepdim=keepdim)
    local_size = tensor.shape[dim]
    lv, li = tensor.max(dim=dim, keepdim=True)
    rank = comm.group_rank(tp)
    li = li + rank * local_size
    all_v = comm.all_gather(lv, dim=gather_dim, group=tp)
    all_i = comm.all_gather(li, dim=gather_dim, group=tp)
    sel = torch.argmax(all_v, dim=dim, keepdim=True)
    out = all_i.gather(dim if dim >= 0 else all_i.dim() - 1, sel)
    if not keepdim:
        out = out.squeeze(dim)
    return out
