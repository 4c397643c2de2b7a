"""Sequence-chunked LM loss.

The (B, S, V) logits of a big-vocab model dominate training activation
memory (llama2-7b at B16/S4096: ~4 GiB bf16 + CE workspaces).  This
computes lm_head + vocab-parallel CE per sequence chunk under
``torch.utils.checkpoint``: each chunk's logits exist only transiently in
forward and are RECOMPUTED in backward — peak logits memory drops from
(S) to (chunk) rows (the reference leans on its compiler's rematerializer
for the same effect)."""

import torch
from torch.utils.checkpoint import checkpoint

from ..parallel.loss_functions import parallel_cross_entropy


def chunked_lm_loss(hidden: torch.Tensor, lm_head, labels: torch.Tensor,
                    num_chunks: int = 4, ignore_index: int = -100):
    """hidden (B, S, H) -> scalar mean-over-valid CE of lm_head(hidden)
    against next-token labels, chunked along S.  ``lm_head`` is the
    (column-parallel) module producing vocab-parallel logits."""
    B, S, _ = hidden.shape
    shift_h = hidden[:, :-1, :]
    shift_l = labels[:, 1:]
    n = shift_h.shape[1]
    bounds = [round(i * n / num_chunks) for i in range(num_chunks + 1)]

    def chunk_loss(h, l):
        return parallel_cross_entropy(lm_head(h).contiguous(), l,
                                      ignore_index=ignore_index).sum()

    total = hidden.new_zeros((), dtype=torch.float32)
    for i in range(num_chunks):
        lo, hi = bounds[i], bounds[i + 1]
        if lo == hi:
            continue
        h = shift_h[:, lo:hi, :].contiguous()
        l = shift_l[:, lo:hi].contiguous()
        total = total + checkpoint(chunk_loss, h, l, use_reentrant=False)
    valid = (shift_l != ignore_index).sum().clamp(min=1)
    return total / valid
