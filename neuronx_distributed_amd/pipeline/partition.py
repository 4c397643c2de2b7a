"""Model partitioning for pipeline parallelism.

Parity with reference ``pipeline/partition.py`` (:18-43 FX split_module,
:280-303 even auto-cuts) and ``pipeline/trace.py`` (:31-39 leaf-module
policy: the parallel layers are traced as LEAVES so their collectives never
enter the graph).
"""

from typing import Dict, List, Optional

import torch
import torch.fx as fx
from torch.fx.passes.split_module import split_module

from ..parallel import ColumnParallelLinear, RowParallelLinear, \
    ParallelEmbedding, GQAQKVColumnParallelLinear
from ..models.llama import RMSNorm, LlamaDecoderLayer, LlamaAttention, LlamaMLP

_DEFAULT_LEAF_MODULES = (
    ColumnParallelLinear, RowParallelLinear, ParallelEmbedding,
    GQAQKVColumnParallelLinear, RMSNorm, LlamaDecoderLayer, LlamaAttention,
    LlamaMLP,
)


class NxDTracer(fx.Tracer):
    """Treats parallel layers (and user-registered classes) as leaves
    (reference pipeline/trace.py:31-39)."""

    def __init__(self, leaf_modules=()):
        super().__init__()
        self.leaf_modules = tuple(leaf_modules) + _DEFAULT_LEAF_MODULES

    def is_leaf_module(self, m, qualname):
        if isinstance(m, self.leaf_modules):
            return True
        return super().is_leaf_module(m, qualname)


def trace_model(model: torch.nn.Module, input_names: Optional[List[str]] = None,
                leaf_modules=()) -> fx.GraphModule:
    """FX symbolic trace (reference pipeline/trace.py:153-219)."""
    tracer = NxDTracer(leaf_modules)
    concrete_args = None
    if input_names is not None:
        import inspect

        sig = inspect.signature(model.forward)
        concrete_args = {
            name: p.default
            for name, p in sig.parameters.items()
            if name not in input_names and p.default is not inspect.Parameter.empty
        }
    graph = tracer.trace(model, concrete_args=concrete_args)
    return fx.GraphModule(model, graph)


def get_auto_pipeline_cuts(model: torch.nn.Module, transformer_layer_cls,
                           num_stages: int) -> List[str]:
    """Evenly distribute transformer layers over stages and return the
    module names starting stages 1..num_stages-1 (reference
    partition.py:280-303)."""
    layer_names = [
        name for name, m in model.named_modules()
        if isinstance(m, transformer_layer_cls)
    ]
    assert len(layer_names) >= num_stages, (
        f"{len(layer_names)} layers < {num_stages} stages")
    per = len(layer_names) / num_stages
    cuts = []
    for s in range(1, num_stages):
        cuts.append(layer_names[int(round(s * per))])
    return cuts


def partition_traced(gm: fx.GraphModule, pipeline_cuts: List[str],
                     num_stages: int) -> fx.GraphModule:
    """split_module with stage assignment by cut-point module names
    (reference partition.py:18-43): a call_module whose qualified name has a
    cut as prefix starts the next stage; every later node stays in >= that
    stage (monotone assignment keeps dataflow forward-only)."""
    cuts = list(pipeline_cuts)
    current = {"stage": 0}

    node_stage: Dict[fx.Node, int] = {}
    stage = 0
    for node in gm.graph.nodes:
        if node.op == "call_module":
            name = node.target
            for i, cut in enumerate(cuts):
                if name == cut or name.startswith(cut + "."):
                    stage = max(stage, i + 1)
        node_stage[node] = stage

    def mod_partition(node):
        return node_stage.get(node, 0)

    split = split_module(gm, None, mod_partition)
    return split


def partition_model(model: torch.nn.Module, num_stages: int,
                    pipeline_cuts: Optional[List[str]] = None,
                    transformer_layer_cls=None, input_names=None,
                    leaf_modules=()):
    """Trace + cut + split; returns (split GraphModule, stage submodules
    list)."""
    gm = trace_model(model, input_names=input_names, leaf_modules=leaf_modules)
    if pipeline_cuts is None:
        assert transformer_layer_cls is not None
        pipeline_cuts = get_auto_pipeline_cuts(model, transformer_layer_cls,
                                               num_stages)
    split = partition_traced(gm, pipeline_cuts, num_stages)
    stages = [getattr(split, f"submod_{i}") for i in range(num_stages)]
    return split, stages
