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


def create_dp_dataloader(dataset, batch_size: int, shuffle: bool = True,
                         seed: int = 0, drop_last: bool = True, **kwargs):
    """DataLoader sharded over the DATA-parallel group (reference
    examples/training create_pretraining_dataset pattern): every TP/PP/CP
    rank inside one DP replica sees the SAME batches (they compute one
    model replica together); different DP ranks see disjoint shards."""
    from torch.utils.data import DataLoader
    from torch.utils.data.distributed import DistributedSampler

    dp = ps.get_data_parallel_size()
    dp_rank = ps.get_data_parallel_rank()
    sampler = DistributedSampler(dataset, num_replicas=dp, rank=dp_rank,
                                 shuffle=shuffle, seed=seed,
                                 drop_last=drop_last)
    return DataLoader(dataset, batch_size=batch_size, sampler=sampler,
                      drop_last=drop_last, **kwargs)


def pad_batch_to_multiple(batch: dict, multiple: int, seq_dim: int = 1,
                          pad_token_id: int = 0, label_pad: int = -100):
    """Right-pad every seq-dim tensor to a multiple (SP needs seq % tp == 0,
    CP seq % cp == 0; reference examples pad in the dataloader).  Labels
    (any key containing 'label') pad with ``label_pad`` so CE ignores the
    padding positions."""
    out = {}
    pad_len = None
    for k, v in batch.items():
        if isinstance(v, torch.Tensor) and v.dim() > seq_dim:
            S = v.shape[seq_dim]
            target = (S + multiple - 1) // multiple * multiple
            pad_len = target - S
            if pad_len == 0:
                out[k] = v
                continue
            fill = label_pad if "label" in k else pad_token_id
            pad_shape = list(v.shape)
            pad_shape[seq_dim] = pad_len
            out[k] = torch.cat(
                [v, v.new_full(pad_shape, fill)], dim=seq_dim)
        else:
            out[k] = v
    return out, (pad_len or 0)
