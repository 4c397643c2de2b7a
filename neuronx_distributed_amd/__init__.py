"""neuronx_distributed_amd — MI355X-native distributed training & inference.

A from-scratch CDNA4 framework with the capabilities and API surface of
aws-neuron/neuronx-distributed: tensor/sequence/pipeline/
context/expert parallelism, ZeRO-1, sharded checkpoints, MoE, LoRA,
quantization and an AOT-style inference builder — on eager PyTorch-ROCm,
hand-written HIP/CDNA4 kernels, and RCCL over xGMI.

Top-level exports mirror the reference's
``src/neuronx_distributed/__init__.py:1-19``.
"""

__version__ = "0.1.0"

from . import parallel

# Reference-compatible alias: `import neuronx_distributed_amd.parallel_layers`
import sys as _sys

parallel_layers = parallel
_sys.modules[__name__ + ".parallel_layers"] = parallel

from . import utils
from . import ops
from . import kernels
from . import operators

# heavier subsystems imported lazily-but-eagerly enough for API parity
from . import optimizer
from . import trainer as _trainer_pkg
from . import pipeline
from . import moe
from . import lora
from . import quantization
from . import inference

# trainer API (reference exports)
from .trainer import (
    neuronx_distributed_config,
    nxd_config,
    initialize_parallel_model,
    initialize_parallel_optimizer,
    save_checkpoint,
    load_checkpoint,
    has_checkpoint,
)
from .trainer.model import NxDModel as NxDTrainModel
from .trainer.optimizer import NxDOptimizer

# inference API (reference exports: ModelBuilder, NxDModel, shard_checkpoint,
# NxDParallelState)
from .inference import (
    ModelBuilder,
    NxDModel,
    NxDParallelState,
    shard_checkpoint,
)

trainer = _trainer_pkg

from .inference.generation import generate
from .inference.speculation import medusa_generate, speculative_generate
