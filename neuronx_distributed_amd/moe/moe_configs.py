"""MoE configuration dataclasses (reference modules/moe/moe_configs.py:
22-292, compacted to the knobs the MI355X build consumes).

The reference's NKI-specific fields (logical_nc_config, block-parallel
kernel selectors, dynamic-while variants) map onto this build's grouped-
GEMM/skip machinery; fields that are pure Neuron-compiler plumbing are
dropped rather than carried as dead weight."""

from dataclasses import dataclass, field
from typing import Callable, Optional

from .model_utils import DEFAULT_BLOCK_SIZE, GLUType


class BlockShardStrategy:
    """Block-parallel sharding strategies (reference K5 variants)."""

    HI_LO = "hi_lo"            # alternate high/low block halves per shard
    PING_PONG = "ping_pong"    # round-robin blocks over shards


@dataclass
class BlockwiseMatmulConfig:
    """Knobs of the blockwise (dropless) expert path (reference :22-133)."""

    block_size: int = DEFAULT_BLOCK_SIZE
    use_torch_block_wise: bool = False   # force the per-expert loop path
    block_sharding_strategy: str = BlockShardStrategy.PING_PONG
    # skip machinery (reference skip_dma): padding-only blocks skip the
    # GEMMs; a static per-expert block budget turns on DROPPING
    skip_dma_token: bool = True
    skip_dma_weight: bool = True
    max_blocks_per_expert: Optional[int] = None
    num_static_blocks: Optional[int] = None
    pad_num_blocks_to_even: bool = False

    @classmethod
    def from_kwargs(cls, **kwargs):
        known = {k: v for k, v in kwargs.items()
                 if k in cls.__dataclass_fields__}
        return cls(**known)


@dataclass
class RoutedExpertsMLPOpsConfig:
    """Geometry + policy of the routed expert MLPs (reference :135-200)."""

    num_experts: int
    top_k: int
    hidden_size: int
    intermediate_size: int
    hidden_act: str = "silu"
    glu_mlp: bool = True
    glu_type: GLUType = GLUType.SWIGLU
    capacity_factor: Optional[float] = None
    normalize_top_k_affinities: bool = False
    early_expert_affinity_modulation: bool = False
    # clamp limits applied around the gate/up projections (reference
    # gate/up_clamp_*_limit); None = no clamping
    gate_clamp_upper_limit: Optional[float] = None
    gate_clamp_lower_limit: Optional[float] = None
    up_clamp_upper_limit: Optional[float] = None
    up_clamp_lower_limit: Optional[float] = None
    init_method: Optional[Callable] = None
    output_layer_init_method: Optional[Callable] = None

    def __post_init__(self):
        if self.top_k > self.num_experts:
            raise ValueError(
                f"top_k ({self.top_k}) > num_experts ({self.num_experts})")
        self.glu_type = GLUType.validate(
            self.glu_type if not isinstance(self.glu_type, GLUType)
            else self.glu_type)


@dataclass
class MoEFusedTKGConfig:
    """Decode-path fusion eligibility knobs (reference moe_fused_tkg)."""

    max_batch: int = 64
    enabled: bool = True
