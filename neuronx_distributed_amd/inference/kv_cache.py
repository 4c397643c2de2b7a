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
            self.k.index_copy_(2, pos, k_new.to(self.k.dtype))
            self.v.index_copy_(2, pos, v_new.to(self.v.dtype))
            return self.k, self.v
        S = k_new.shape[2]
        self.k[:, :, pos:pos + S] = k_new
        self.v[:, :, pos:pos + S] = v_new
        return self.k[:, :, :pos + S], self.v[:, :, :pos + S]


def build_kv_caches(n_layers: int, batch: int, n_kv_heads_local: int,
                    max_seq: int, head_dim: int, dtype=torch.bfloat16,
                    device=None, window=None) -> List[KVCache]:
    """``window``: sliding-window models get RollingKVCache (O(window)
    memory per layer) when the window is smaller than max_seq."""
    if window is not None and window < max_seq:
        return [RollingKVCache(batch, n_kv_heads_local, int(window),
                               head_dim, dtype, device)
                for _ in range(n_layers)]
    return [KVCache(batch, n_kv_heads_local, max_seq, head_dim, dtype, device)
            for _ in range(n_layers)]


class RollingKVCache(KVCache):
    """Window-bounded cache for sliding-window (Mistral-style) models:
    ``window`` slots addressed ``pos % window``, with ``slot_pos`` (W,)
    int64 tracking each slot's GLOBAL position (-1 = unwritten) so the
    decode step masks by true position while memory stays O(window)
    regardless of context length.  hipGraph-compatible: the tensor-pos
    update is index_copy_ on both the buffers and slot_pos.

    Supported pattern: one-shot prefill from position 0, then per-token
    decode (the serving pattern); chunked prefill at pos > 0 would need
    a rotated gather and raises."""

    def __init__(self, batch: int, n_kv_heads: int, window: int,
                 head_dim: int, dtype=torch.bfloat16, device=None):
        super().__init__(batch, n_kv_heads, window, head_dim, dtype, device)
        self.window = window
        self.slot_pos = torch.full((window,), -1, dtype=torch.long,
                                   device=self.k.device)

    def position_index(self) -> torch.Tensor:
        return self.slot_pos

    def update(self, k_new: torch.Tensor, v_new: torch.Tensor, pos):
        if isinstance(pos, torch.Tensor):  # decode: one token
            slot = pos % self.window
            self.k.index_copy_(2, slot, k_new.to(self.k.dtype))
            self.v.index_copy_(2, slot, v_new.to(self.v.dtype))
            self.slot_pos.index_copy_(0, slot, pos)
            return self.k, self.v
        S = k_new.shape[2]
        if pos != 0:
            raise NotImplementedError(
                "RollingKVCache supports one-shot prefill from pos 0")
        # keep only the last ``window`` prefill rows (older ones can
        # never be attended again)
        tail = min(S, self.window)
        idx = torch.arange(S - tail, S, device=self.k.device)
        slot = idx % self.window
        self.k.index_copy_(2, slot, k_new[:, :, S - tail:].to(self.k.dtype))
        self.v.index_copy_(2, slot, v_new[:, :, S - tail:].to(self.v.dtype))
        self.slot_pos.index_copy_(0, slot, idx)
        # prefill attention runs over the CURRENT chunk directly (the
        # cache held nothing before position 0)
        return k_new, v_new
