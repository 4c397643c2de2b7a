"""Calibration observers (reference quantization/observer.py:12).

Attach to float layers, run calibration batches, then convert with the
observed activation/weight ranges (static quantization)."""

from typing import Dict

import torch
import torch.nn as nn


class MinMaxObserver(nn.Module):
    """Tracks the running |x| max of whatever passes through."""

    def __init__(self):
        super().__init__()
        self.register_buffer("amax", torch.zeros(()))
        self.observed = False

    @torch.no_grad()
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        m = x.detach().abs().amax().float().cpu()
        self.amax = torch.maximum(self.amax, m)
        self.observed = True
        return x

    def scale(self, qmax: float) -> torch.Tensor:
        return (self.amax.clamp(min=1e-8) / qmax)


def attach_observers(model: nn.Module, layer_types) -> Dict[str, MinMaxObserver]:
    """Register a MinMaxObserver on the INPUT of every matching layer via
    forward-pre hooks; returns name -> observer."""
    observers: Dict[str, MinMaxObserver] = {}
    for name, mod in model.named_modules():
        if isinstance(mod, layer_types):
            obs = MinMaxObserver()
            observers[name] = obs

            def hook(m, args, _obs=obs):
                if args and isinstance(args[0], torch.Tensor):
                    _obs(args[0])

            mod.register_forward_pre_hook(hook)
    return observers


def collect_activation_scales(observers: Dict[str, MinMaxObserver],
                              qmax: float = 448.0
                              ) -> Dict[str, torch.Tensor]:
    return {n: o.scale(qmax) for n, o in observers.items() if o.observed}
