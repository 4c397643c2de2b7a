"""Model-construction helpers: meta-device init + sequential materialization.

Parity with reference ``utils/model_utils.py`` (:257-358): ``init_on_device``
context manager (build on meta to avoid host OOM), ``get_model_sequential``
(materialize + move ranks in groups with rendezvous), shared-weight re-tie
(:48-82).
"""

import contextlib

import torch
import torch.nn as nn

from ..parallel import comm, parallel_state as ps


@contextlib.contextmanager
def init_on_device(device: torch.device, include_buffers: bool = True):
    """Patch nn.Module.register_parameter/buffer so construction happens on
    ``device`` (meta for deferred init; reference model_utils.py:257-333)."""
    old_register_parameter = nn.Module.register_parameter
    old_register_buffer = nn.Module.register_buffer

    def register_empty_parameter(module, name, param):
        old_register_parameter(module, name, param)
        if param is not None:
            param_cls = type(module._parameters[name])
            kwargs = module._parameters[name].__dict__
            extra = {
                k: v for k, v in kwargs.items()
                if k in ("tensor_model_parallel", "partition_dim",
                         "partition_stride", "num_partitions", "kv_shared",
                         "expert_model_parallel", "sequence_parallel_enabled")
            }
            new = param_cls(module._parameters[name].to(device),
                            requires_grad=param.requires_grad)
            for k, v in extra.items():
                setattr(new, k, v)
            module._parameters[name] = new

    def register_empty_buffer(module, name, buffer, persistent=True):
        old_register_buffer(module, name, buffer, persistent)
        if buffer is not None:
            module._buffers[name] = module._buffers[name].to(device)

    try:
        nn.Module.register_parameter = register_empty_parameter
        if include_buffers:
            nn.Module.register_buffer = register_empty_buffer
        yield
    finally:
        nn.Module.register_parameter = old_register_parameter
        if include_buffers:
            nn.Module.register_buffer = old_register_buffer


def reinit_model(model: nn.Module, device: torch.device, param_init_fn=None):
    """Materialize a meta-built model on ``device``."""
    model.to_empty(device=device)
    if param_init_fn is not None:
        for m in model.modules():
            param_init_fn(m)
    else:
        for m in model.modules():
            if hasattr(m, "reset_parameters"):
                m.reset_parameters()
    return model


def get_model_sequential(model_fn, device: torch.device,
                         sequential_move_factor: int = 11,
                         param_init_fn=None) -> nn.Module:
    """Build + move rank models to device in groups of
    ``sequential_move_factor`` ranks with a rendezvous between groups
    (host-RAM OOM control; reference model_utils.py:335-358)."""
    world = torch.distributed.get_world_size() if torch.distributed.is_initialized() else 1
    rank = torch.distributed.get_rank() if torch.distributed.is_initialized() else 0
    model = None
    for group_start in range(0, world, sequential_move_factor):
        if group_start <= rank < group_start + sequential_move_factor:
            model = model_fn()
            model = model.to(device)
        comm.barrier()
    return model


def retie_shared_weights(model: nn.Module, shared_weight_names) -> None:
    """Re-tie weights that aliasing broke during materialization
    (reference model_utils.py:48-82).  ``shared_weight_names`` is a list of
    (src_path, dst_path) dotted attribute pairs."""
    for src, dst in shared_weight_names:
        src_mod, src_attr = _resolve(model, src)
        dst_mod, dst_attr = _resolve(model, dst)
        setattr(dst_mod, dst_attr, getattr(src_mod, src_attr))


def _resolve(model, path):
    parts = path.split(".")
    mod = model
    for p in parts[:-1]:
        mod = getattr(mod, p)
    return mod, parts[-1]


def is_hf_pretrained_model(model) -> bool:
    for klass in type(model).__mro__:
        if klass.__module__.startswith("transformers") and \
                klass.__name__.endswith("PreTrainedModel"):
            return True
    return False
