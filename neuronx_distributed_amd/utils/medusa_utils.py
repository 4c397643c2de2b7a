"""Medusa speculative-decoding buffers (reference utils/medusa_utils.py):
tree attention masks + candidate gathering for multi-head draft decoding."""

from typing import List

import torch


def generate_medusa_buffers(medusa_choices: List[List[int]], device=None):
    """Build the tree-attention buffers from a Medusa choice tree (each
    entry is a path of per-head top-k indices, e.g. [0], [0,1], [1,0]...).

    Returns dict with: medusa_attn_mask (T,T), tree_indices (T),
    medusa_position_ids (T), retrieve_indices (L, depth+1)."""
    sorted_choices = sorted(medusa_choices, key=lambda p: (len(p), p))
    n = len(sorted_choices) + 1  # + root

    attn_mask = torch.eye(n, dtype=torch.bool)
    attn_mask[:, 0] = True
    paths = {(): 0}
    for i, path in enumerate(sorted_choices):
        paths[tuple(path)] = i + 1
        for d in range(1, len(path)):
            anc = paths[tuple(path[:d])]
            attn_mask[i + 1, anc] = True

    position_ids = torch.tensor([0] + [len(p) for p in sorted_choices])

    # tree_indices: which draft-token candidate feeds each node:
    # head (depth-1), top-k index path[-1]
    max_k = max((p[-1] for p in sorted_choices), default=0) + 1
    tree_indices = torch.zeros(n, dtype=torch.long)
    for i, path in enumerate(sorted_choices):
        tree_indices[i + 1] = (len(path) - 1) * max_k + path[-1] + 1

    # retrieve_indices: leaf-to-root paths for candidate extraction
    leaves = [p for p in sorted_choices
              if not any(len(q) == len(p) + 1 and q[:len(p)] == list(p)
                         for q in sorted_choices)]
    depth = max((len(p) for p in sorted_choices), default=0)
    retrieve = torch.full((len(leaves), depth + 1), -1, dtype=torch.long)
    for li, leaf in enumerate(leaves):
        retrieve[li, 0] = 0
        for d in range(1, len(leaf) + 1):
            retrieve[li, d] = paths[tuple(leaf[:d])]

    out = {
        "medusa_attn_mask": attn_mask,
        "tree_indices": tree_indices,
        "medusa_position_ids": position_ids,
        "retrieve_indices": retrieve,
    }
    if device is not None:
        out = {k: v.to(device) for k, v in out.items()}
    return out


class MedusaHead(torch.nn.Module):
    """One residual-MLP draft head (reference medusa head design)."""

    def __init__(self, hidden_size: int, vocab_size: int, dtype=None):
        super().__init__()
        from ..parallel.layers import ColumnParallelLinear

        self.proj = torch.nn.Linear(hidden_size, hidden_size, dtype=dtype)
        self.act = torch.nn.SiLU()
        self.lm_head = ColumnParallelLinear(hidden_size, vocab_size,
                                            bias=False, gather_output=False,
                                            dtype=dtype)

    def forward(self, hidden):
        return self.lm_head(hidden + self.act(self.proj(hidden)))
