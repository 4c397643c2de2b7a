"""Parallel layers — the TP/SP/DP/PP/CP/EP substrate.

API parity with the reference's ``neuronx_distributed.parallel_layers``
(``parallel_layers/__init__.py:1-37``).
"""

from . import parallel_state
from . import comm
from . import mappings
from . import layers
from . import grads
from . import loss_functions
from . import random
from . import utils
from . import checkpointing
from . import pad

from .parallel_state import (
    initialize_model_parallel,
    model_parallel_is_initialized,
    destroy_model_parallel,
    get_tensor_model_parallel_group,
    get_tensor_model_parallel_rank,
    get_tensor_model_parallel_size,
    get_data_parallel_group,
    get_data_parallel_rank,
    get_data_parallel_size,
    get_pipeline_model_parallel_group,
    get_pipeline_model_parallel_rank,
    get_pipeline_model_parallel_size,
    rmsg,
)
from .layers import (
    ColumnParallelLinear,
    RowParallelLinear,
    ParallelEmbedding,
    LinearWithAsyncCommunication,
    linear_with_async_allreduce,
    SPMDRank,
)
from .qkv_linear import GQAQKVColumnParallelLinear
from .layer_norm import LayerNorm
from .loss_functions import parallel_cross_entropy, from_parallel_logits_to_logprobs
from .mappings import (
    copy_to_tensor_model_parallel_region,
    gather_from_tensor_model_parallel_region,
    reduce_from_tensor_model_parallel_region,
    scatter_to_tensor_model_parallel_region,
    scatter_to_sequence_parallel_region,
    gather_from_sequence_parallel_region,
    reduce_scatter_to_sequence_parallel_region,
    enter_expert_parallel_region,
    exit_expert_parallel_region,
)
from .grads import (
    clip_grad_norm,
    get_grad_norm,
    bucket_allreduce_gradients,
    allreduce_sequence_parallel_gradients,
    allreduce_context_parallel_gradients,
)
from .random import model_parallel_manual_seed, get_rng_state_tracker
from .checkpointing import save, load

# Lists consumed by the pipeline tracer (reference parallel_layers/__init__.py
# PARALLEL_MODULES / PARALLEL_FUNCTIONS, pipeline/trace.py:176-187)
PARALLEL_MODULES = [
    ColumnParallelLinear,
    RowParallelLinear,
    ParallelEmbedding,
    GQAQKVColumnParallelLinear,
]
PARALLEL_FUNCTIONS = [
    parallel_cross_entropy,
    copy_to_tensor_model_parallel_region,
    gather_from_tensor_model_parallel_region,
    reduce_from_tensor_model_parallel_region,
    scatter_to_tensor_model_parallel_region,
]

from .conv import OutputChannelParallelConv2d, InputChannelParallelConv2d
PARALLEL_MODULES.append(OutputChannelParallelConv2d)
PARALLEL_MODULES.append(InputChannelParallelConv2d)
