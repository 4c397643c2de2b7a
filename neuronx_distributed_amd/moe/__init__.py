from .model import MoE
from .routing import RouterTopK, RouterSinkhorn, GroupLimitedRouter
from .expert_mlps import ExpertMLPs
from .moe_parallel_layers import (
    ExpertFusedColumnParallelLinear,
    ExpertFusedRowParallelLinear,
)
from .shared_experts import SharedExperts
from .loss_function import load_balancing_loss_func
from .token_shuffling import token_shuffle, token_unshuffle
from .moe_fused_tkg import MoEFusedTKG
