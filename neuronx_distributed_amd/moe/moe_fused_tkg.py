"""Whole-MoE-layer fused token-generation path (reference
modules/moe/moe_fused_tkg.py:24-250 ``MoEFusedTKG``).

The reference fuses RMSNorm + router + expert MLPs + shared experts +
residual into ONE NKI kernel for decode (seq_len == 1).  The MI355X
equivalent composes the already-fused HIP kernels — ops.rmsnorm, a skinny
router GEMM, ops.moe_decode_glu (gather + gate/up + SwiGLU + down +
affinity-scatter in two kernels), the shared-expert fused gate-up — into
one module whose whole body is hipGraph-capturable, so a decode step pays
the same ~4 kernel dispatches the reference's monolithic kernel does.

Eligibility mirrors the reference (:142-179): decode batch <= 64, GLU
activation, kernels available; everything else falls back to the unfused
router/experts/shared submodules (:209-272).
"""

from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..parallel import parallel_state as ps
from ..parallel.mappings import reduce_from_tensor_model_parallel_region
from .expert_mlps import ExpertMLPs
from .routing import RouterBase
from .shared_experts import SharedExperts


class MoEFusedTKG(nn.Module):
    """Decode-path facade: ``forward(hidden (B,1,H)) -> (B,1,H)`` applying
    norm -> router -> experts (+ shared experts) -> residual."""

    MAX_FUSED_BATCH = 64  # reference eligibility :142-179

    def __init__(self, router: RouterBase, expert_mlps: ExpertMLPs,
                 norm: Optional[nn.Module] = None,
                 shared_experts: Optional[SharedExperts] = None):
        super().__init__()
        self.router = router
        self.expert_mlps = expert_mlps
        self.norm = norm
        self.shared_experts = shared_experts

    def can_fuse(self, hidden: torch.Tensor) -> bool:
        return (not isinstance(hidden, torch.fx.Proxy)
                and not self.training
                and hidden.is_cuda and hidden.dtype == torch.bfloat16
                and hidden.shape[0] * hidden.shape[1] <= self.MAX_FUSED_BATCH
                and self.expert_mlps.glu_mlp
                and self.expert_mlps.ep_size == 1
                and ops.is_available() and ops.moe_decode_available())

    def _fused(self, hidden):
        B, S, H = hidden.shape
        h = hidden.reshape(-1, H)
        normed = ops.rmsnorm(h, self.norm.weight,
                             self.norm.variance_epsilon) \
            if self.norm is not None else h
        _, aff, idx = self.router(normed)
        out = ops.moe_decode_glu(normed, self.expert_mlps.gate_up_proj.weight,
                                 self.expert_mlps.down_proj.weight,
                                 aff, idx)
        out = reduce_from_tensor_model_parallel_region(out)
        if self.shared_experts is not None:
            out = out + self.shared_experts(normed)
        return (h + out).reshape(B, S, H)  # fused residual

    def _unfused(self, hidden):
        B, S, H = hidden.shape
        h = hidden.reshape(-1, H)
        normed = self.norm(h) if self.norm is not None else h
        _, aff, idx = self.router(normed)
        out = self.expert_mlps(normed, aff, idx)
        out = reduce_from_tensor_model_parallel_region(out)
        if self.shared_experts is not None:
            out = out + self.shared_experts(normed)
        return (h + out).reshape(B, S, H)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        if self.can_fuse(hidden):
            return self._fused(hidden)
        return self._unfused(hidden)
