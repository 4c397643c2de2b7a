"""Shared experts (reference modules/moe/shared_experts.py:73-207):
always-on dense MLP beside the routed experts.

Two weight placements, mirroring the reference:
* default — TP-sharded gate/up (column) + down (row), output TP-partial
  until the MoE layer's delayed reduce;
* ``replicate_for_sp=True`` — weights fully REPLICATED on every rank
  (reference :118-126 per-rank singleton "TP groups"): under
  sequence-parallel context encoding each rank already owns a distinct
  token slice, so replicated weights need NO TP collective at all.  The
  output is then divided by tp so the MoE layer's unconditional delayed
  all-reduce reconstructs the exact sum.
"""

import torch
import torch.nn as nn

from .. import ops
from ..parallel import parallel_state as ps
from ..parallel.layers import ColumnParallelLinear, RowParallelLinear


class SharedExperts(nn.Module):
    def __init__(self, hidden_size: int, intermediate_size: int,
                 num_shared_experts: int = 1, dtype=None, device=None,
                 init_method=None, fused_gate_up: bool = True,
                 replicate_for_sp: bool = False):
        super().__init__()
        I = intermediate_size * num_shared_experts
        self.replicate_for_sp = replicate_for_sp
        dtype = dtype or torch.get_default_dtype()
        if replicate_for_sp:
            # full (non-sharded) weights on every rank; deterministic init
            # matches the sharded variant's master weights
            init_method = init_method or (
                lambda t: nn.init.normal_(t, std=0.02))
            self.gate_up_proj = nn.Parameter(torch.empty(
                2 * I, hidden_size, dtype=dtype, device=device))
            self.down_proj = nn.Parameter(torch.empty(
                hidden_size, I, dtype=dtype, device=device))
            if device != torch.device("meta"):
                with torch.no_grad():
                    for w in (self.gate_up_proj, self.down_proj):
                        m = torch.empty(w.shape, dtype=torch.float32,
                                        device="cpu")
                        init_method(m)
                        w.data.copy_(m.to(dtype))
            return
        self.gate_up_proj = ColumnParallelLinear(
            hidden_size, 2 * I, bias=False, gather_output=False, stride=2,
            dtype=dtype, device=device, init_method=init_method)
        self.down_proj = RowParallelLinear(
            I, hidden_size, bias=False, input_is_parallel=True, dtype=dtype,
            device=device, init_method=init_method)

    def _act(self, gu):
        if gu.dtype == torch.bfloat16 and gu.is_cuda:
            return ops.swiglu(gu)
        I = gu.shape[-1] // 2
        return torch.nn.functional.silu(gu[..., :I]) * gu[..., I:]

    def forward(self, x):
        if self.replicate_for_sp:
            # every rank computes the FULL shared MLP on its own (SP-local)
            # tokens — no TP collective (reference :118-126); the MoE layer
            # adds this AFTER its delayed TP reduce of the routed output
            gu = torch.nn.functional.linear(x, self.gate_up_proj)
            return torch.nn.functional.linear(self._act(gu), self.down_proj)
        return self.down_proj(self._act(self.gate_up_proj(x)))
