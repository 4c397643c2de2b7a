"""Tensor-parallel LoRA adapters (reference modules/lora/tp_layer.py:15,62).

For ColumnParallelLinear the B factor is sharded on the output dim (A
replicated); for RowParallelLinear the A factor is sharded on the input dim
(B replicated) — the low-rank bottleneck stays replicated so no extra
collectives are needed beyond the base layer's."""

import math

import torch
import torch.nn as nn

from ..parallel.layers import ColumnParallelLinear, RowParallelLinear
from ..parallel.qkv_linear import GQAQKVColumnParallelLinear
from ..parallel.utils import set_tensor_model_parallel_attributes
from ..parallel import parallel_state as ps
from .layer import LoraLayerBase


class LoraParallelLinear(LoraLayerBase):
    def __init__(self, base, rank=16, alpha=32.0, dropout=0.0):
        super().__init__(base, rank, alpha, dropout)
        dtype = base.weight.dtype
        dev = base.weight.device
        if isinstance(base, ColumnParallelLinear):
            in_f = base.input_size
            out_local = base.output_size_per_partition
            self.lora_A = nn.Parameter(torch.zeros(rank, in_f, dtype=dtype,
                                                   device=dev))
            self.lora_B = nn.Parameter(torch.zeros(out_local, rank,
                                                   dtype=dtype, device=dev))
            set_tensor_model_parallel_attributes(
                self.lora_B, ps.get_tensor_model_parallel_size() > 1, 0)
        elif isinstance(base, RowParallelLinear):
            in_local = base.input_size_per_partition
            out_f = base.output_size
            self.lora_A = nn.Parameter(torch.zeros(rank, in_local,
                                                   dtype=dtype, device=dev))
            set_tensor_model_parallel_attributes(
                self.lora_A, ps.get_tensor_model_parallel_size() > 1, 1)
            self.lora_B = nn.Parameter(torch.zeros(out_f, rank, dtype=dtype,
                                                   device=dev))
        else:
            raise TypeError(type(base))
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))

    def forward(self, x):
        from ..parallel.mappings import (
            gather_from_sequence_parallel_region,
            reduce_from_tensor_model_parallel_region,
            reduce_scatter_to_sequence_parallel_region,
        )

        out = self.base_layer(x)
        xd = self.lora_dropout(x)
        if isinstance(self.base_layer, RowParallelLinear):
            # input is TP-sharded; A sharded to match -> the rank-space
            # bottleneck is a TP-partial sum that must be reduced (the
            # base layer's own reduce already happened inside it)
            bott = xd @ self.lora_A.t()
            if self.base_layer.sequence_parallel_enabled:
                bott = reduce_scatter_to_sequence_parallel_region(bott,
                                                                  seq_dim=0)
            else:
                bott = reduce_from_tensor_model_parallel_region(bott)
            lora = bott @ self.lora_B.t()
        else:
            from ..parallel.mappings import (
                copy_to_tensor_model_parallel_region)

            if getattr(self.base_layer, "sequence_parallel_enabled", False):
                xd = gather_from_sequence_parallel_region(
                    xd, seq_dim=0, to_model_parallel=True)
            # lora_A is REPLICATED while lora_B is output-sharded: the true
            # dL/dA is the SUM of rank-local contributions, so the
            # bottleneck gets identity-fwd / all-reduce-bwd
            bott = copy_to_tensor_model_parallel_region(xd @ self.lora_A.t())
            lora = bott @ self.lora_B.t()
        if isinstance(out, tuple):
            return (out[0] + lora * self.scaling,) + out[1:]
        return out + lora * self.scaling

    @torch.no_grad()
    def merge(self):
        self.base_layer.weight += (self.lora_B @ self.lora_A) * self.scaling
        return self.base_layer


class LoraGQAQKVParallelLinear(LoraLayerBase):
    """Adapters on the fused q/k/v projections (reference tp_layer.py:62)."""

    def __init__(self, base: GQAQKVColumnParallelLinear, rank=16, alpha=32.0,
                 dropout=0.0):
        super().__init__(base, rank, alpha, dropout)
        dtype = base.weight_q.dtype
        dev = base.weight_q.device
        in_f = base.weight_q.shape[1]
        self.lora_A = nn.Parameter(torch.zeros(rank, in_f, dtype=dtype,
                                               device=dev))
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
        for name, w in (("q", base.weight_q), ("k", base.weight_k),
                        ("v", base.weight_v)):
            B = nn.Parameter(torch.zeros(w.shape[0], rank, dtype=dtype,
                                         device=dev))
            set_tensor_model_parallel_attributes(
                B, ps.get_tensor_model_parallel_size() > 1, 0)
            setattr(self, f"lora_B_{name}", B)

    def forward(self, x):
        from ..parallel.mappings import gather_from_sequence_parallel_region

        q, k, v = self.base_layer(x)
        xd = self.lora_dropout(x)
        if self.base_layer.sequence_parallel_enabled:
            xd = gather_from_sequence_parallel_region(
                xd, seq_dim=0, to_model_parallel=True)
        from ..parallel.mappings import copy_to_tensor_model_parallel_region

        # replicated A + sharded B_q/k/v: all-reduce the bottleneck grad
        bott = copy_to_tensor_model_parallel_region(xd @ self.lora_A.t())
        q = q + (bott @ self.lora_B_q.t()) * self.scaling
        k = k + (bott @ self.lora_B_k.t()) * self.scaling
        v = v + (bott @ self.lora_B_v.t()) * self.scaling
        return q, k, v

    @torch.no_grad()
    def merge(self):
        self.base_layer.weight_q += (self.lora_B_q @ self.lora_A) * self.scaling
        self.base_layer.weight_k += (self.lora_B_k @ self.lora_A) * self.scaling
        self.base_layer.weight_v += (self.lora_B_v @ self.lora_A) * self.scaling
        return self.base_layer
