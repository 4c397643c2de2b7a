"""Expert MLP execution strategies (reference modules/moe/expert_mlps_v2.py).

Strategies (dispatch policy :1407-1499):
* ``forward_all_experts`` (:366-483): every expert processes every token,
  affinity-masked — used for decode and small token counts.
* ``forward_capacity_factor`` (:484-593): capacity
  ``C = min(T, ceil(T*k*cf/E))``; position-in-expert via masked cumsum;
  over-capacity tokens dropped; gather -> fused 3-D MLP -> unpermute with
  affinity scaling.  Static-shaped (hipGraph-friendly).
* blockwise grouped-GEMM (K4/K5) arrives with the HIP kernel; until then
  capacity-factor/all-experts cover its configs.

EP: training wraps the expert compute in enter/exit_expert_parallel_region
all-to-alls (reference experts.py:182-213).
"""

import math
from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..parallel import parallel_state as ps
from ..parallel.mappings import (
    copy_to_tensor_model_parallel_region,
    enter_expert_parallel_region,
    exit_expert_parallel_region,
)
from ..parallel.utils import divide
from .moe_parallel_layers import (
    ExpertFusedColumnParallelLinear,
    ExpertFusedRowParallelLinear,
)


class ExpertMLPs(nn.Module):
    SELECTIVE_LOADING_THRESHOLD = 1.0

    def __init__(self, num_experts: int, hidden_size: int,
                 intermediate_size: int, top_k: int,
                 capacity_factor: Optional[float] = None,
                 glu_mlp: bool = True, dtype=None, device=None,
                 init_method=None):
        super().__init__()
        self.num_experts = num_experts
        self.hidden_size = hidden_size
        self.intermediate_size = intermediate_size
        self.top_k = top_k
        self.capacity_factor = capacity_factor
        self.glu_mlp = glu_mlp

        ep = ps.get_expert_model_parallel_size()
        self.ep_size = ep
        self.num_experts_local = divide(num_experts, ep)
        self.ep_rank = ps.get_expert_model_parallel_rank()

        # fused [gate; up] column weights + down row weights
        out_mult = 2 if glu_mlp else 1
        self.gate_up_proj = ExpertFusedColumnParallelLinear(
            self.num_experts_local, hidden_size,
            out_mult * intermediate_size, dtype=dtype, device=device,
            init_method=init_method, stride=out_mult)
        self.down_proj = ExpertFusedRowParallelLinear(
            self.num_experts_local, intermediate_size, hidden_size,
            dtype=dtype, device=device, init_method=init_method,
            reduce_output=False)

    # -- core fused 3-D MLP ------------------------------------------------
    def _mlp(self, x):
        """x (E_local, C, H) -> (E_local, C, H); TP-partial output (the MoE
        layer does the delayed reduce, reference moe/model.py:224-245)."""
        gu = self.gate_up_proj(x)
        if self.glu_mlp:
            I = gu.shape[-1] // 2
            act = ops.swiglu(gu) if gu.dtype == torch.bfloat16 and gu.is_cuda \
                else torch.nn.functional.silu(gu[..., :I]) * gu[..., I:]
        else:
            act = torch.nn.functional.gelu(gu)
        return self.down_proj(act)

    # -- strategies --------------------------------------------------------
    def forward_all_experts(self, hidden, expert_affinities):
        """(T,H), (T,E) -> (T,H): every (local) expert runs all tokens.
        With EP, tokens are all-gathered over the EP group, local experts
        process the union, and the result is summed back via
        reduce-scatter (reference forward_all_experts_EP,
        expert_mlps_v2.py:394).  The collectives are autograd-aware so EP
        training gradients flow back to the tokens and the router."""
        from ..parallel.mappings import (gather_from_group,
                                         reduce_scatter_to_group)

        if self.ep_size > 1:
            hidden = gather_from_group(hidden, dim=0, group_name="ep")
            expert_affinities = gather_from_group(expert_affinities, dim=0,
                                                  group_name="ep")
        T = hidden.shape[0]
        x = hidden.unsqueeze(0).expand(self.num_experts_local, T,
                                       self.hidden_size)
        out = self._mlp(x)  # (E_local, T, H)
        e0 = self.ep_rank * self.num_experts_local
        aff = expert_affinities[:, e0:e0 + self.num_experts_local]  # (T,El)
        out = torch.einsum("eth,te->th", out.float(), aff.float()).to(hidden.dtype)
        if self.ep_size > 1:
            out = reduce_scatter_to_group(out, dim=0, group_name="ep")
        return out

    def forward_capacity_factor(self, hidden, expert_affinities, expert_index):
        """reference expert_mlps_v2.py:484-593."""
        T = hidden.shape[0]
        E = self.num_experts
        k = self.top_k
        C = min(T, math.ceil(T * k * self.capacity_factor / E))

        # (T,k) expert assignment -> one-hot (T,k,E)
        onehot = torch.nn.functional.one_hot(expert_index, E)  # (T,k,E)
        # position of each token within its expert via cumsum over tokens
        flat = onehot.reshape(T * k, E)
        positions = flat.cumsum(dim=0) - 1  # (T*k, E)
        pos_in_expert = (positions * flat).sum(-1).reshape(T, k)  # (T,k)
        keep = (pos_in_expert < C) & (expert_index >= 0)

        # scatter token hidden into (E, C, H)
        expert_inputs = hidden.new_zeros(E, C, self.hidden_size)
        tok_idx = torch.arange(T, device=hidden.device).unsqueeze(1).expand(T, k)
        e_flat = expert_index[keep]
        p_flat = pos_in_expert[keep]
        t_flat = tok_idx[keep]
        expert_inputs[e_flat, p_flat] = hidden[t_flat]

        if self.ep_size > 1:
            if self.training:
                expert_inputs = enter_expert_parallel_region(expert_inputs)
                local = expert_inputs.reshape(self.num_experts_local, -1,
                                              self.hidden_size)
            else:
                e0 = self.ep_rank * self.num_experts_local
                local = expert_inputs[e0:e0 + self.num_experts_local]
        else:
            local = expert_inputs

        out_local = self._mlp(local)

        if self.ep_size > 1 and self.training:
            # (E/ep, ep*C, H) -> (E, C, H)
            expert_out = exit_expert_parallel_region(out_local)
        elif self.ep_size > 1:
            full = out_local.new_zeros(E, C, self.hidden_size)
            e0 = self.ep_rank * self.num_experts_local
            full[e0:e0 + self.num_experts_local] = out_local
            expert_out = full  # world-reduce happens in the MoE layer
        else:
            expert_out = out_local

        # unpermute with affinity scaling; dropped tokens contribute 0
        aff = expert_affinities.gather(-1, expert_index)  # (T,k)
        out = hidden.new_zeros(T, self.hidden_size).float()
        gathered = expert_out[e_flat, p_flat].float()
        out.index_add_(0, t_flat,
                       gathered * aff[keep].unsqueeze(-1).float())
        return out.to(hidden.dtype)

    def forward_selective(self, hidden, expert_affinities, expert_index,
                          normalize_top_k_affinities: bool = False):
        """Selective loading (reference expert_mlps_v2.py:595-689): at
        token generation with few tokens, gather ONLY the top-k experts'
        weight slices per token (T*k slices instead of all E experts) and
        run them as one batched GEMM — vectorized over (token, k) pairs
        instead of the reference's per-token python loop."""
        T = hidden.shape[0]
        k = self.top_k
        idx = expert_index.reshape(-1)  # (T*k,) local expert ids
        aff = expert_affinities.gather(-1, expert_index)  # (T,k)
        if normalize_top_k_affinities:
            aff = torch.nn.functional.normalize(aff, p=1.0, dim=1)

        x = hidden.unsqueeze(1).expand(T, k, self.hidden_size)
        x = x.reshape(T * k, 1, self.hidden_size)
        gu = self.gate_up_proj(x, expert_indices=idx)  # (T*k, 1, 2I/tp)
        if self.glu_mlp:
            I = gu.shape[-1] // 2
            act = ops.swiglu(gu) if gu.dtype == torch.bfloat16 and gu.is_cuda \
                else torch.nn.functional.silu(gu[..., :I]) * gu[..., I:]
        else:
            act = torch.nn.functional.gelu(gu)
        out = self.down_proj(act, expert_indices=idx)  # (T*k, 1, H)
        out = out.reshape(T, k, self.hidden_size).float()
        return (out * aff.unsqueeze(-1).float()).sum(dim=1).to(hidden.dtype)

    def forward_blockwise(self, hidden, expert_affinities, expert_index):
        """No-drop blockwise strategy (reference expert_mlps_v2.py:691 +
        blockwise.py K4): fixed-size expert blocks, grouped GEMMs."""
        from .blockwise import blockwise_mm, compute_block_indices, \
            DEFAULT_BLOCK_SIZE

        assert self.ep_size == 1, "blockwise with EP: use capacity_factor"
        block_size = min(DEFAULT_BLOCK_SIZE,
                         max(32, hidden.shape[0] // 4))
        tpi, b2e, _ = compute_block_indices(
            expert_index, self.num_experts, block_size,
            max_blocks_per_expert=getattr(self, "blockwise_dropping_blocks",
                                          None))
        return blockwise_mm(hidden, expert_affinities,
                            self.gate_up_proj.weight.to(hidden.dtype),
                            self.down_proj.weight.to(hidden.dtype),
                            tpi, b2e, expert_index, block_size,
                            glu=self.glu_mlp)

    def forward(self, hidden, expert_affinities, expert_index):
        """Dispatch (reference :1407-1499): training -> capacity_factor if
        set (>0); <=0 -> blockwise; None -> all-experts.  Small inference
        batches take the fused HIP decode path (K9): per-expert slot
        blocks, weights streamed once, gather/SwiGLU/scatter fused."""
        hidden = copy_to_tensor_model_parallel_region(hidden)
        traced = isinstance(hidden, torch.fx.Proxy)
        if (not traced and not self.training
                and self.glu_mlp and self.ep_size == 1 and hidden.is_cuda
                and hidden.dtype == torch.bfloat16
                and self.gate_up_proj.weight.dtype == torch.bfloat16
                and hidden.shape[0] <= 512 and ops.moe_decode_available()):
            return ops.moe_decode_glu(hidden, self.gate_up_proj.weight,
                                      self.down_proj.weight,
                                      expert_affinities, expert_index)
        if (not traced and not self.training and self.ep_size == 1
                and hidden.shape[0] * self.top_k
                < self.SELECTIVE_LOADING_THRESHOLD * self.num_experts):
            # token generation with few expert hits: load only the top-k
            # experts' weights (reference dispatch :1407-1499)
            return self.forward_selective(hidden, expert_affinities,
                                          expert_index)
        if self.capacity_factor is not None and self.capacity_factor > 0:
            return self.forward_capacity_factor(hidden, expert_affinities,
                                                expert_index)
        if self.capacity_factor is not None and self.capacity_factor <= 0 \
                and self.ep_size == 1:
            return self.forward_blockwise(hidden, expert_affinities,
                                          expert_index)
        return self.forward_all_experts(hidden, expert_affinities)
