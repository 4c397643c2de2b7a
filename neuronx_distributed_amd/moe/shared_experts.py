"""Shared experts (reference modules/moe/shared_experts.py:73-207):
always-on dense MLP beside the routed experts, TP-sharded gate/up/down."""

import torch
import torch.nn as nn

from .. import ops
from ..parallel.layers import ColumnParallelLinear, RowParallelLinear


class SharedExperts(nn.Module):
    def __init__(self, hidden_size: int, intermediate_size: int,
                 num_shared_experts: int = 1, dtype=None, device=None,
                 init_method=None, fused_gate_up: bool = True):
        super().__init__()
        I = intermediate_size * num_shared_experts
        self.gate_up_proj = ColumnParallelLinear(
            hidden_size, 2 * I, bias=False, gather_output=False, stride=2,
            dtype=dtype, device=device, init_method=init_method)
        self.down_proj = RowParallelLinear(
            I, hidden_size, bias=False, input_is_parallel=True, dtype=dtype,
            device=device, init_method=init_method)

    def forward(self, x):
        gu = self.gate_up_proj(x)
        if gu.dtype == torch.bfloat16 and gu.is_cuda:
            act = ops.swiglu(gu)
        else:
            I = gu.shape[-1] // 2
            act = torch.nn.functional.silu(gu[..., :I]) * gu[..., I:]
        return self.down_proj(act)
