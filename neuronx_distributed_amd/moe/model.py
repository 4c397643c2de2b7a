"""MoE layer facade (reference modules/moe/model.py:14, fwd :154-303).

Composes router + expert MLPs (+ optional shared experts); the TP
all-reduce of the expert output is DELAYED to after affinity scaling
(reference :224-245), matching the RowParallel ``reduce_output=False``
design."""

from typing import Optional

import torch
import torch.nn as nn

from ..parallel import parallel_state as ps
from ..parallel.mappings import (
    reduce_from_tensor_model_parallel_region,
    reduce_scatter_to_sequence_parallel_region,
    scatter_to_sequence_parallel_region,
)
from .expert_mlps import ExpertMLPs
from .routing import RouterBase
from .shared_experts import SharedExperts
from .token_shuffling import token_shuffle, token_unshuffle


class MoE(nn.Module):
    def __init__(self, router: RouterBase, expert_mlps: ExpertMLPs,
                 shared_experts: Optional[SharedExperts] = None,
                 return_router_logits: bool = True,
                 sequence_parallel_enabled: bool = False,
                 token_shuffle_group_size: int = 1):
        super().__init__()
        self.router = router
        self.expert_mlps = expert_mlps
        self.shared_experts = shared_experts
        self.return_router_logits = return_router_logits
        self.sequence_parallel_enabled = sequence_parallel_enabled
        self.token_shuffle_group_size = token_shuffle_group_size

    def forward(self, hidden_states: torch.Tensor):
        """hidden (B,S,H) (or (S/tp,B,H) under SP) -> same shape (+ router
        logits when requested)."""
        orig_shape = hidden_states.shape
        h = hidden_states.reshape(-1, orig_shape[-1])  # (T,H)

        shuffle_perm = None
        if self.token_shuffle_group_size > 1 and self.training:
            h, shuffle_perm = token_shuffle(h)

        router_logits, expert_affinities, expert_index = self.router(h)
        out = self.expert_mlps(h, expert_affinities, expert_index)

        # delayed TP reduce (reference :224-245)
        out = reduce_from_tensor_model_parallel_region(out)

        if self.shared_experts is not None:
            out = out + self.shared_experts(h)

        if shuffle_perm is not None:
            out = token_unshuffle(out, shuffle_perm)

        out = out.reshape(orig_shape)
        if self.return_router_logits:
            return out, router_logits
        return out
