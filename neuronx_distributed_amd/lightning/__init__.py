"""PyTorch-Lightning integration (reference lightning/ 1,264 LoC:
NeuronXLAStrategy, NeuronLTModule, NeuronCheckpointIO).

Lightning is an optional dependency — everything here import-guards it and
raises a clear error when absent (this image ships without lightning; the
classes are exercised by duck-typed unit tests).

Deliberately NOT ported from the reference (XLA-substrate glue that is
unnecessary on ROCm): ``accelerator.py`` / ``launcher.py`` /
``precision_plugin.py`` exist there to teach Lightning about XLA devices,
``xmp.spawn`` process launch and XLA bf16 autocast — on MI355X the stock
CUDA accelerator, torchrun launcher and native precision plugins work
as-is; ``progress_bar.py`` / ``logger.py`` / ``neuron_hooks_callback.py``
paper over xm.mark_step-driven metric staleness, which eager execution
does not have."""

from .strategy import NeuronLTStrategy
from .module import NeuronLTModule
from .checkpoint_io import NeuronCheckpointIO
