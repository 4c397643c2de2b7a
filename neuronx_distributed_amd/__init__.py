"""neuronx_distributed_amd — MI355X-native distributed training & inference.

A from-scratch CDNA4 framework with the capabilities and API surface of
aws-neuron/neuronx-distributed: tensor/sequence/pipeline/
context/expert parallelism, ZeRO-1, sharded checkpoints, MoE, LoRA,
quantization and an AOT-style inference builder — on eager PyTorch-ROCm,
hand-written HIP/CDNA4 kernels, and RCCL over xGMI.
"""

__version__ = "0.1.0"

from . import parallel
# Reference-compatible alias: `import neuronx_distributed_amd.parallel_layers`
import sys as _sys

parallel_layers = parallel
_sys.modules[__name__ + ".parallel_layers"] = parallel

from . import utils
