"""Intermediate-tensor capture for numerical debugging (reference
utils/tensor_capture/api.py:16-95: registry + forward-hook capture)."""

from typing import Dict, List, Optional

import torch
import torch.nn as nn

_CAPTURED: Dict[str, torch.Tensor] = {}
_HOOKS = []


def enable_tensor_capture(model: nn.Module,
                          module_names: Optional[List[str]] = None):
    """Install forward hooks capturing the outputs of the named modules
    (all leaf modules when None)."""
    disable_tensor_capture()

    def make_hook(name):
        def hook(mod, inputs, output):
            out = output[0] if isinstance(output, tuple) else output
            if isinstance(out, torch.Tensor):
                _CAPTURED[name] = out.detach()
        return hook

    for name, mod in model.named_modules():
        if module_names is None or name in module_names:
            _HOOKS.append(mod.register_forward_hook(make_hook(name)))
    return model


def get_captured_tensors() -> Dict[str, torch.Tensor]:
    return dict(_CAPTURED)


def disable_tensor_capture():
    global _HOOKS
    for h in _HOOKS:
        h.remove()
    _HOOKS = []
    _CAPTURED.clear()
