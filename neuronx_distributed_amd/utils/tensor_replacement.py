"""Tensor replacement registry (reference utils/tensor_replacement/):
swap named modules' outputs (or inputs) with provided tensors during
forward — the counterpart of tensor_capture for fault-injection and
debug-divergence experiments."""

from typing import Any, Dict

import torch
import torch.nn as nn


class TensorReplacer:
    """Replace the OUTPUT of named submodules with fixed tensors (or the
    result of a callable receiving the original output)."""

    def __init__(self, model: nn.Module):
        self.model = model
        self._handles = []
        self._replacements: Dict[str, Any] = {}

    def replace(self, module_name: str, value) -> "TensorReplacer":
        self._replacements[module_name] = value
        return self

    def __enter__(self):
        for name, mod in self.model.named_modules():
            if name in self._replacements:
                val = self._replacements[name]

                def hook(m, args, out, _val=val):
                    return _val(out) if callable(_val) else _val

                self._handles.append(mod.register_forward_hook(hook))
        return self

    def __exit__(self, *exc):
        for h in self._handles:
            h.remove()
        self._handles = []
        return False
