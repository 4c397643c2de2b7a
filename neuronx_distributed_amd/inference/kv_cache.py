"""KV cache for GQA decode (reference trace/ StateInitializer-aliased
INPUT_STATE tensors, nxd_model.py:123-139 — here plain device-resident
buffers updated in place, hipGraph-compatible)."""

from typing import List

import torch


class KVCache:
    """One layer's cache: k/v (B, Hkv_local, S_max, D) bf16."""

    def __init__(self, batch: int, n_kv_heads: int, max_seq: int,
                 head_dim: int, dtype=torch.bfloat16, device=None):
        device = device or (torch.device("cuda") if torch.cuda.is_available()
                            else "cpu")
        self.k = torch.zeros(batch, n_kv_heads, max_seq, head_dim,
                             dtype=dtype, device=device)
        self.v = torch.zeros_like(self.k)
        self.max_seq = max_seq

    def update(self, k_new: torch.Tensor, v_new: torch.Tensor, pos):
        """k_new (B,H,S,D) written at [pos:pos+S]; returns views of the
        cache covering [0:pos+S].

        ``pos`` may be a device int64 tensor of shape (1,) (hipGraph
        decode): the write becomes index_copy_ and the FULL buffers are
        returned (attention masks by position — no dynamic shapes)."""
        if isinstance(pos, torch.Tensor):
            self.k.index_copy_(2, pos, k_new)
            self.v.index_copy_(2, pos, v_new)
            return self.k, self.v
        S = k_new.shape[2]
        self.k[:, :, pos:pos + S] = k_new
        self.v[:, :, pos:pos + S] = v_new
        return self.k[:, :, :pos + S], self.v[:, :, :pos + S]


def build_kv_caches(n_layers: int, batch: int, n_kv_heads_local: int,
                    max_seq: int, head_dim: int, dtype=torch.bfloat16,
                    device=None) -> List[KVCache]:
    return [KVCache(batch, n_kv_heads_local, max_seq, head_dim, dtype, device)
            for _ in range(n_layers)]
