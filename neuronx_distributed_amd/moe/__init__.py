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
from .model_utils import ACT2FN, ACTFunc, GLUType, DEFAULT_BLOCK_SIZE
from .moe_configs import (BlockwiseMatmulConfig, BlockShardStrategy,
                          MoEFusedTKGConfig, RoutedExpertsMLPOpsConfig)
from .moe_config_validator import MoeConfigValidator
from .moe_process_group import (init_tensor_expert_parallel_moe_process_groups,
                                get_moe_tp_ep_group, get_moe_ep_group,
                                destroy_moe_model_parallel)
