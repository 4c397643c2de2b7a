"""Batch slicing for context parallelism (reference utils/batch_utils.py:19):
each CP rank takes a contiguous S/cp slice of the sequence; position ids
offset by cp_rank * (S/cp) (reference modeling_llama_nxd.py:608-616)."""

import torch

from ..parallel import parallel_state as ps


def get_batch_on_this_context_parallel_rank(batch: dict, seq_dim: int = 1):
    cp = ps.get_context_model_parallel_size()
    if cp == 1:
        return batch, 0
    r = ps.get_context_model_parallel_rank()
    out = {}
    chunk_len = None
    for k, v in batch.items():
        if isinstance(v, torch.Tensor) and v.dim() > seq_dim:
            assert v.shape[seq_dim] % cp == 0
            out[k] = v.chunk(cp, dim=seq_dim)[r].contiguous()
            chunk_len = v.shape[seq_dim] // cp
        else:
            out[k] = v
    return out, r * (chunk_len or 0)
