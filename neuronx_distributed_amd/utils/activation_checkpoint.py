"""Activation checkpointing (reference utils/activation_checkpoint.py:31-84:
wraps given module classes with the torch checkpoint fn).  On MI355X we use
torch.utils.checkpoint with non-reentrant mode (plays well with RCCL async
collectives and the flash kernels' saved tensors)."""

from functools import partial
from typing import Iterable, Optional, Type, Union

import torch
import torch.nn as nn
from torch.utils.checkpoint import checkpoint


class CheckpointWrapper(nn.Module):
    def __init__(self, module: nn.Module):
        super().__init__()
        self._checkpoint_wrapped_module = module

    @property
    def module(self):
        return self._checkpoint_wrapped_module

    def forward(self, *args, **kwargs):
        return checkpoint(self._checkpoint_wrapped_module, *args,
                          use_reentrant=False, **kwargs)

    def named_parameters(self, *args, **kwargs):
        return self._checkpoint_wrapped_module.named_parameters(*args, **kwargs)

    def state_dict(self, *args, **kwargs):
        return self._checkpoint_wrapped_module.state_dict(*args, **kwargs)

    def load_state_dict(self, *args, **kwargs):
        return self._checkpoint_wrapped_module.load_state_dict(*args, **kwargs)


def apply_activation_checkpointing(model: nn.Module,
                                   check_fn=None,
                                   activation_checkpoint_classes=None) -> None:
    """Wrap matching submodules in-place (reference trainer.py:201-232
    'full' = every decoder-layer class)."""
    if activation_checkpoint_classes is not None:
        classes = tuple(activation_checkpoint_classes)
        check_fn = lambda m: isinstance(m, classes)  # noqa: E731
    assert check_fn is not None

    def _wrap(parent):
        for name, child in parent.named_children():
            if isinstance(child, CheckpointWrapper):
                continue
            if check_fn(child):
                setattr(parent, name, CheckpointWrapper(child))
            else:
                _wrap(child)

    _wrap(model)
