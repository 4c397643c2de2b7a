"""Token shuffling for MoE load balance (reference
modules/moe/token_shuffling.py:64-117): random permutation + all-to-all
over the token-shuffle group before routing, inverted after."""

import torch

from ..parallel import comm, parallel_state as ps


def token_shuffle(hidden: torch.Tensor, seed: int = None):
    """(T,H) -> shuffled (T,H); returns (shuffled, permutation)."""
    T = hidden.shape[0]
    if seed is not None:
        g = torch.Generator(device=hidden.device)
        g.manual_seed(seed)
        perm = torch.randperm(T, device=hidden.device, generator=g)
    else:
        perm = torch.randperm(T, device=hidden.device)
    h = hidden[perm]
    if "token_shuffle" in ps._GROUPS:
        h = comm.all_to_all(h, 0, 0, group=ps.get_group_info("token_shuffle"))
    return h, perm


def token_unshuffle(hidden: torch.Tensor, perm: torch.Tensor):
    if "token_shuffle" in ps._GROUPS:
        hidden = comm.all_to_all(hidden, 0, 0,
                                 group=ps.get_group_info("token_shuffle"))
    inv = torch.empty_like(perm)
    inv[perm] = torch.arange(perm.numel(), device=perm.device)
    return hidden[inv]
