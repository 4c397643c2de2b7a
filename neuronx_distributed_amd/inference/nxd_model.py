"""Runtime inference model with shape-bucket routing + hipGraph capture.

Parity with reference ``trace/nxd_model/nxd_model.py:41,460`` (multi-bucket
router keyed on input shapes; state tensors/KV cache held module-side) —
the ``torch.classes.neuron.SPMDModel`` C++ runtime becomes "eager module +
one captured hipGraph per bucket" (SURVEY §3.4 MI355X translation):
``torch.cuda.CUDAGraph`` on ROCm IS hipGraph.
"""

from typing import Any, Dict, Optional, Tuple

import torch
import torch.nn as nn

from ..utils.logger import get_logger

logger = get_logger(__name__)


class NxDModel(nn.Module):
    def __init__(self, module: nn.Module, use_hip_graphs: bool = True):
        super().__init__()
        self.module = module
        self.use_hip_graphs = use_hip_graphs and torch.cuda.is_available()
        self._buckets: Dict[Tuple, str] = {}       # shape key -> tag
        self._graphs: Dict[str, Any] = {}          # tag -> (graph, in, out)

    @staticmethod
    def _shape_key(kwargs):
        return tuple(sorted(
            (k, tuple(v.shape)) for k, v in kwargs.items()
            if isinstance(v, torch.Tensor)))

    def add_bucket(self, tag: str, example_inputs: Dict[str, Any]):
        """Register a shape bucket (reference nxd_model.py:87,170-190)."""
        self._buckets[self._shape_key(example_inputs)] = tag

    def route(self, kwargs) -> Optional[str]:
        return self._buckets.get(self._shape_key(kwargs))

    # -- deployment artifact (reference nxd_model.py:709,924 torchscript
    # export; MI355X: weights + bucket manifest, graphs re-capture lazily) --

    def save(self, path: str) -> None:
        """Write a deployment artifact: ``weights.safetensors`` (deduped,
        shared tensors cloned) + ``manifest.pt`` with the bucket routing
        table.  ``NxDModel.load(path, model_fn)`` reconstructs it; the
        hipGraphs re-capture on first use (they are device-state, not
        serializable)."""
        import os

        from ..utils.safetensors_utils import save_safetensors

        os.makedirs(path, exist_ok=True)
        save_safetensors(self.module.state_dict(),
                         os.path.join(path, "weights.safetensors"))
        torch.save({
            "buckets": self._buckets,
            "use_hip_graphs": self.use_hip_graphs,
        }, os.path.join(path, "manifest.pt"))

    @classmethod
    def load(cls, path: str, model_fn, strict: bool = True,
             **model_kwargs) -> "NxDModel":
        """Rebuild from a ``save()`` artifact: ``model_fn(**model_kwargs)``
        constructs the architecture; weights and the bucket table load
        from the artifact."""
        import os

        from ..utils.safetensors_utils import load_safetensors

        module = model_fn(**model_kwargs)
        sd = load_safetensors(os.path.join(path, "weights.safetensors"))
        module.load_state_dict(sd, strict=strict)
        if torch.cuda.is_available():
            module = module.cuda()
        module.eval()
        manifest = torch.load(os.path.join(path, "manifest.pt"),
                              map_location="cpu", weights_only=False)
        model = cls(module, use_hip_graphs=manifest["use_hip_graphs"])
        model._buckets = manifest["buckets"]
        return model

    @torch.no_grad()
    def _capture(self, tag: str, kwargs):
        """Capture one hipGraph for this bucket: static input buffers are
        reused on every call (KV-cache state lives in module buffers and
        mutates inside the graph)."""
        static_in = {k: (v.clone() if isinstance(v, torch.Tensor) else v)
                     for k, v in kwargs.items()}
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self.module(**static_in)
        torch.cuda.current_stream().wait_stream(s)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            static_out = self.module(**static_in)
        self._graphs[tag] = (graph, static_in, static_out)
        logger.info("captured hipGraph for bucket '%s'", tag)

    @torch.no_grad()
    def forward(self, **kwargs):
        # accept host inputs: route tensors to the module's device (the
        # reference SPMDModel runtime also owns the host->device copy)
        try:
            dev = next(self.module.parameters()).device
            kwargs = {k: (v.to(dev) if isinstance(v, torch.Tensor)
                          and v.device != dev else v)
                      for k, v in kwargs.items()}
        except StopIteration:
            pass
        tag = self.route(kwargs)
        if not self.use_hip_graphs or tag is None:
            return self.module(**kwargs)
        if tag not in self._graphs:
            self._capture(tag, kwargs)
        graph, static_in, static_out = self._graphs[tag]
        for k, v in kwargs.items():
            if isinstance(v, torch.Tensor):
                static_in[k].copy_(v)
        graph.replay()
        if isinstance(static_out, torch.Tensor):
            return static_out.clone()
        return type(static_out)(o.clone() if isinstance(o, torch.Tensor)
                                else o for o in static_out)


class StateInitializer:
    """KV-cache / state tensor factory (reference
    trace/nxd_model/base_nxd_model.py:11)."""

    def __init__(self, shapes: Dict[str, tuple], dtypes: Dict[str, Any],
                 device=None):
        self.shapes = shapes
        self.dtypes = dtypes
        self.device = device or (
            torch.device("cuda") if torch.cuda.is_available() else "cpu")

    def __call__(self):
        return {
            name: torch.zeros(shape, dtype=self.dtypes[name],
                              device=self.device)
            for name, shape in self.shapes.items()
        }
