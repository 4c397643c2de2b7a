"""MoE routers (reference modules/moe/routing.py).

``RouterTopK`` (:155-210), ``RouterSinkhorn`` (:213-314),
``GroupLimitedRouter`` (DeepSeek-V3 no-aux-loss, :316-458).  Semantics
preserved: router logits computed in fp32 (the reference uses fp64 to
dodge XLA bf16 auto-downcast, :116-126 — eager ROCm has no auto-downcast,
fp32 suffices and is what the math needs), Sinkhorn runs a FIXED iteration
count in fp32 under no_grad (:224,283-314).
"""


import torch
import torch.nn as nn
import torch.nn.functional as F

from ..parallel import parallel_state as ps
from ..parallel.mappings import gather_from_sequence_parallel_region


class RouterBase(nn.Module):
    def __init__(self, num_experts: int, top_k: int, hidden_size: int,
                 act_fn: str = "softmax", dtype=torch.float32,
                 sequence_parallel_enabled: bool = False):
        super().__init__()
        self.num_experts = num_experts
        self.top_k = top_k
        self.hidden_size = hidden_size
        self.act_fn = act_fn
        self.sequence_parallel_enabled = sequence_parallel_enabled
        self.linear_router = nn.Linear(hidden_size, num_experts, bias=False,
                                       dtype=torch.float32)

    def _logits(self, hidden: torch.Tensor) -> torch.Tensor:
        # (T,H) -> (T,E) in fp32
        if self.sequence_parallel_enabled:
            # gather sharded sequence so every rank routes the full token set
            hidden = gather_from_sequence_parallel_region(
                hidden, seq_dim=0, to_model_parallel=False)
        return self.linear_router(hidden.float())

    def _activate(self, logits: torch.Tensor) -> torch.Tensor:
        if self.act_fn == "softmax":
            return torch.softmax(logits, dim=-1)
        if self.act_fn == "sigmoid":
            return torch.sigmoid(logits)
        raise ValueError(self.act_fn)


class RouterTopK(RouterBase):
    """Top-k routing (reference routing.py:155-210)."""

    def __init__(self, *args, apply_act_fn_over_topk: bool = False, **kwargs):
        super().__init__(*args, **kwargs)
        self.apply_act_fn_over_topk = apply_act_fn_over_topk

    def forward(self, hidden: torch.Tensor):
        T = hidden.shape[0] * (hidden.shape[1] if hidden.dim() == 3 else 1)
        h = hidden.reshape(-1, self.hidden_size)
        logits = self._logits(h)
        if self.apply_act_fn_over_topk:
            # select by raw logits, normalize only over the selected k
            vals, idx = torch.topk(logits, self.top_k, dim=-1)
            weights = torch.softmax(vals, dim=-1)
            affinities = torch.zeros_like(logits).scatter(-1, idx, weights)
        else:
            probs = self._activate(logits)
            vals, idx = torch.topk(probs, self.top_k, dim=-1)
            affinities = torch.zeros_like(probs).scatter(-1, idx, vals)
        return logits, affinities, idx


class RouterSinkhorn(RouterBase):
    """Sinkhorn-balanced top-1 routing (reference routing.py:213-314):
    a CONSTANT number of normalization iterations (static-graph decision we
    keep for hipGraph capture), fp32, no_grad; affinities come from the
    activated logits, selection from the Sinkhorn-normalized matrix."""

    def __init__(self, *args, sinkhorn_iterations: int = 30, **kwargs):
        super().__init__(*args, **kwargs)
        self.sinkhorn_iterations = sinkhorn_iterations
        assert self.top_k == 1, "RouterSinkhorn is top-1 (reference :224)"

    @torch.no_grad()
    def _sinkhorn(self, cost: torch.Tensor) -> torch.Tensor:
        d0 = torch.ones(cost.size(0), device=cost.device, dtype=cost.dtype)
        d1 = torch.ones(cost.size(1), device=cost.device, dtype=cost.dtype)
        eps = 1e-8
        cost = torch.exp(cost)
        for _ in range(self.sinkhorn_iterations):
            d0 = (1.0 / d0.size(0)) / ((cost * d1.unsqueeze(0)).sum(1) + eps)
            d1 = (1.0 / d1.size(0)) / ((cost * d0.unsqueeze(1)).sum(0) + eps)
        return cost * d1.unsqueeze(0) * d0.unsqueeze(1)

    def forward(self, hidden: torch.Tensor):
        h = hidden.reshape(-1, self.hidden_size)
        logits = self._logits(h)
        if self.training:
            balanced = self._sinkhorn(logits.detach().float())
            _, idx = torch.topk(balanced, 1, dim=-1)
        else:
            _, idx = torch.topk(logits, 1, dim=-1)
        probs = self._activate(logits)
        vals = probs.gather(-1, idx)
        affinities = torch.zeros_like(probs).scatter(-1, idx, vals)
        return logits, affinities, idx


class GroupLimitedRouter(RouterBase):
    """DeepSeek-V3-style group-limited no-aux-loss routing (reference
    routing.py:316-458): sigmoid scores + e_score_correction_bias; experts
    grouped into n_groups; group score = sum of its top-2 member scores
    (:415-426); only experts inside the top-``topk_group`` groups are
    eligible (:391-413); affinities renormalized over the chosen k and
    scaled by routed_scaling_factor."""

    def __init__(self, num_experts, top_k, hidden_size, n_groups: int = 8,
                 topk_group: int = 4, routed_scaling_factor: float = 1.0,
                 norm_topk_prob: bool = True, **kwargs):
        kwargs.setdefault("act_fn", "sigmoid")
        super().__init__(num_experts, top_k, hidden_size, **kwargs)
        assert num_experts % n_groups == 0
        self.n_groups = n_groups
        self.topk_group = topk_group
        self.routed_scaling_factor = routed_scaling_factor
        self.norm_topk_prob = norm_topk_prob
        self.e_score_correction_bias = nn.Parameter(
            torch.zeros(num_experts, dtype=torch.float32),
            requires_grad=False)

    def forward(self, hidden: torch.Tensor):
        h = hidden.reshape(-1, self.hidden_size)
        logits = self._logits(h)
        scores = torch.sigmoid(logits)
        scores_for_choice = scores + self.e_score_correction_bias

        T = scores.shape[0]
        grouped = scores_for_choice.view(T, self.n_groups, -1)
        gk = min(2, grouped.shape[-1])
        group_scores = grouped.topk(gk, dim=-1)[0].sum(dim=-1)  # (T, G)
        group_idx = torch.topk(group_scores, self.topk_group, dim=-1)[1]
        group_mask = torch.zeros_like(group_scores).scatter(-1, group_idx, 1.0)
        expert_mask = group_mask.unsqueeze(-1).expand(
            T, self.n_groups, self.num_experts // self.n_groups).reshape(T, -1)
        masked_scores = scores_for_choice.masked_fill(expert_mask == 0,
                                                      float("-inf"))
        _, idx = torch.topk(masked_scores, self.top_k, dim=-1)
        vals = scores.gather(-1, idx)
        if self.norm_topk_prob:
            vals = vals / (vals.sum(dim=-1, keepdim=True) + 1e-20)
        vals = vals * self.routed_scaling_factor
        affinities = torch.zeros_like(scores).scatter(-1, idx, vals)
        return logits, affinities, idx
