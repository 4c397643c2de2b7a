"""Token shuffling for MoE load balance (reference
modules/moe/token_shuffling.py:64-117): random permutation + all-to-all
over the token-shuffle group before routing, inverted after (BASE Layers,
arXiv:2103.16716 — tokens within one worker's sequence are correlated, so
shuffling spreads them across workers for better expert balance).

The all-to-all is AUTOGRAD-AWARE (reference _AllToAllForTokenShuffle):
token shuffling runs in training, so gradients must ride the inverse
all-to-all back to the producing rank.
"""

import torch

from ..parallel import parallel_state as ps
from ..parallel.mappings import all_to_all as _autograd_all_to_all
from ..parallel.mappings import (
    gather_from_sequence_parallel_region,
    scatter_to_sequence_parallel_region,
)


def _shuffle_all_to_all(hidden: torch.Tensor) -> torch.Tensor:
    if "token_shuffle" not in ps._GROUPS or \
            ps.get_token_shuffle_group_size() == 1:
        return hidden
    return _autograd_all_to_all(hidden, 0, 0, group_name="token_shuffle")


def all_to_all_for_shuffle(hidden: torch.Tensor,
                           input_is_sequence_parallel: bool = True):
    """reference token_shuffling.py all_to_all_for_shuffle: non-SP inputs
    are scattered to sequence-parallel form around the exchange so each
    rank trades equal slices."""
    if not input_is_sequence_parallel:
        hidden = scatter_to_sequence_parallel_region(hidden, seq_dim=0)
    hidden = _shuffle_all_to_all(hidden)
    if not input_is_sequence_parallel:
        hidden = gather_from_sequence_parallel_region(
            hidden, seq_dim=0, to_model_parallel=False)
    return hidden


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
    return _shuffle_all_to_all(h), perm


def token_unshuffle(hidden: torch.Tensor, perm: torch.Tensor):
    # the equal-split all-to-all is self-inverse
    hidden = _shuffle_all_to_all(hidden)
    inv = torch.empty_like(perm)
    inv[perm] = torch.arange(perm.numel(), device=perm.device)
    return hidden[inv]
