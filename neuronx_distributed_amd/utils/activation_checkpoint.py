"""Activation checkpointing (reference utils/activation_checkpoint.py:31-84:
wraps given module classes with the torch checkpoint fn).  On MI355X we use
torch.utils.checkpoint with non-reentrant mode (plays well with RCCL async
collectives and the flash kernels' saved tensors)."""


import torch
import torch.nn as nn
from torch.utils.checkpoint import checkpoint


_PREFIX = "_checkpoint_wrapped_module."


class CheckpointWrapper(nn.Module):
    def __init__(self, module: nn.Module):
        super().__init__()
        self._checkpoint_wrapped_module = module
        # keep state-dict keys identical to the unwrapped module
        self._register_state_dict_hook(self._strip_prefix_hook)
        self._register_load_state_dict_pre_hook(self._add_prefix_hook,
                                                with_module=True)

    @staticmethod
    def _strip_prefix_hook(module, state_dict, prefix, local_metadata):
        for key in list(state_dict.keys()):
            if key.startswith(prefix + _PREFIX):
                new = prefix + key[len(prefix) + len(_PREFIX):]
                state_dict[new] = state_dict.pop(key)
        return state_dict

    @staticmethod
    def _add_prefix_hook(module, state_dict, prefix, *args):
        for key in list(state_dict.keys()):
            if key.startswith(prefix) and not key.startswith(prefix + _PREFIX):
                new = prefix + _PREFIX + key[len(prefix):]
                state_dict[new] = state_dict.pop(key)

    @property
    def module(self):
        return self._checkpoint_wrapped_module

    def forward(self, *args, **kwargs):
        return checkpoint(self._checkpoint_wrapped_module, *args,
                          use_reentrant=False, **kwargs)


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
