"""PyTorch-Lightning integration (reference lightning/ 1,264 LoC:
NeuronXLAStrategy, NeuronLTModule, NeuronCheckpointIO).

Lightning is an optional dependency — everything here import-guards it and
raises a clear error when absent (this image ships without lightning; the
classes are exercised by duck-typed unit tests)."""

from .strategy import NeuronLTStrategy
from .module import NeuronLTModule
from .checkpoint_io import NeuronCheckpointIO
