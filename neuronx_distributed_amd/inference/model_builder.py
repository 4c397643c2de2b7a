"""ModelBuilder (reference trace/model_builder_v2.py:33 + v1 :441).

API-parity inference builder: register shape buckets via .trace(), then
.compile() produces an :class:`NxDModel` wrapping the eager module with
hipGraph capture per bucket.  ``shard_checkpoint`` lives in functions.py.
"""

from typing import Any, Callable, Dict, Optional

import torch

from .functions import TraceArtifact, trace as _trace, compile as _compile
from .nxd_model import NxDModel


class ModelBuilder:
    def __init__(self, router: Optional[Callable] = None, tp_degree: int = 1,
                 checkpoint_loader: Optional[Callable] = None,
                 compiler_workdir: Optional[str] = None, debug: bool = False,
                 model_cls=None, model_fn: Optional[Callable] = None,
                 model_kwargs: Optional[dict] = None):
        self.tp_degree = tp_degree
        self.checkpoint_loader = checkpoint_loader
        self.model_fn = model_fn or model_cls
        self.model_kwargs = model_kwargs or {}
        self.artifacts: Dict[str, TraceArtifact] = {}
        self._module = None

    def trace(self, example_inputs: Dict[str, Any], tag: str = "default"):
        self.artifacts[tag] = _trace(self.model_fn, example_inputs, tag,
                                     self.tp_degree)
        return self

    def compile(self, priority_model_key: Optional[str] = None,
                use_hip_graphs: bool = True) -> NxDModel:
        """Build the module once (optionally loading a checkpoint) and wrap
        with bucket routing; hipGraphs capture lazily per bucket on GPU."""
        module = self.model_fn(**self.model_kwargs)
        if self.checkpoint_loader is not None:
            sd = self.checkpoint_loader()
            module.load_state_dict(sd, strict=False)
        if torch.cuda.is_available():
            module = module.cuda()
        module.eval()
        nxd = NxDModel(module, use_hip_graphs=use_hip_graphs)
        for tag, art in self.artifacts.items():
            _compile(art)
            nxd.add_bucket(tag, art.example_inputs)
        self._module = module
        return nxd
