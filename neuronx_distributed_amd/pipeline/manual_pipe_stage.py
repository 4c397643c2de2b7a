"""Manual (non-traced) pipeline partition (reference
pipeline/manual_pipe_stage.py:14 ``PipelineStageModule``): the user supplies
an ordered layer list; layers are distributed evenly over stages; shared
weights are registered by name."""

from typing import Callable, Dict, List, Optional

import torch
import torch.nn as nn


class PipelineStageModule(nn.Module):
    def __init__(self, layers: List[nn.Module], num_stages: int,
                 stage_index: int,
                 partition_fn: Optional[Callable] = None):
        super().__init__()
        self.num_stages = num_stages
        self.stage_index = stage_index
        if partition_fn is not None:
            assignment = partition_fn(layers, num_stages)
        else:
            assignment = self._even_partition(layers, num_stages)
        self.stage_modules = nn.ModuleList(
            [m for m, s in zip(layers, assignment) if s == stage_index])
        self._shared_weights: Dict[str, nn.Parameter] = {}

    @staticmethod
    def _even_partition(layers, num_stages):
        n = len(layers)
        per = n / num_stages
        return [min(int(i / per), num_stages - 1) for i in range(n)]

    def register_shared_weight(self, name: str, param: nn.Parameter):
        self._shared_weights[name] = param

    @property
    def shared_weights(self):
        return self._shared_weights

    def forward(self, *args):
        out = args
        for m in self.stage_modules:
            out = m(*out) if isinstance(out, tuple) else m(out)
        return out
