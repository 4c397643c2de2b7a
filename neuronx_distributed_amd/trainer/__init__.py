from .trainer import (
    neuronx_distributed_config,
    nxd_config,
    initialize_parallel_model,
    initialize_parallel_optimizer,
)
from .model import NxDModel
from .optimizer import NxDOptimizer
from .checkpoint import save_checkpoint, load_checkpoint, has_checkpoint
