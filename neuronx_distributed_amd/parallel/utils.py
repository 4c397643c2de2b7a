"""Shard math + TP parameter attribute tagging.

Parity with reference ``parallel_layers/utils.py`` (divide/split helpers
:1-54, attribute tagging :55-88, autocast :202-227) — the
``tensor_model_parallel``/``partition_dim``/``partition_stride`` attributes
are part of the public sharded-checkpoint contract.
"""

from typing import Any, Optional

import torch

_TP_ATTRS = {
    "tensor_model_parallel": False,
    "partition_dim": -1,
    "partition_stride": 1,
    "num_partitions": 1,
}

EXPERT_PARALLEL_ATTR = "expert_model_parallel"


def ensure_divisibility(numerator: int, denominator: int) -> None:
    if numerator % denominator != 0:
        raise ValueError(f"{numerator} is not divisible by {denominator}")


def divide(numerator: int, denominator: int) -> int:
    ensure_divisibility(numerator, denominator)
    return numerator // denominator


def set_tensor_model_parallel_attributes(param: torch.Tensor, is_parallel: bool,
                                         dim: int, stride: int = 1,
                                         num_partitions: int = 1) -> None:
    for attr in _TP_ATTRS:
        assert not hasattr(param, attr) or getattr(param, "tensor_model_parallel") in (
            None, False, True
        )
    param.tensor_model_parallel = is_parallel
    param.partition_dim = dim
    param.partition_stride = stride
    param.num_partitions = num_partitions


def copy_tensor_model_parallel_attributes(dst: torch.Tensor, src: torch.Tensor) -> None:
    for attr in _TP_ATTRS:
        if hasattr(src, attr):
            setattr(dst, attr, getattr(src, attr))


def set_defaults_if_not_set_tensor_model_parallel_attributes(param) -> None:
    for attr, default in _TP_ATTRS.items():
        if not hasattr(param, attr):
            setattr(param, attr, default)


def param_is_tensor_parallel(param) -> bool:
    return getattr(param, "tensor_model_parallel", False)


def param_is_expert_parallel(param) -> bool:
    return getattr(param, EXPERT_PARALLEL_ATTR, False)


def split_tensor_along_dim(tensor: torch.Tensor, dim: int, num_partitions: int,
                           contiguous_split_chunks: bool = False):
    size = divide(tensor.shape[dim], num_partitions)
    chunks = torch.split(tensor, size, dim=dim)
    if contiguous_split_chunks:
        return tuple(c.contiguous() for c in chunks)
    return chunks


def split_tensor_along_last_dim(tensor, num_partitions,
                                contiguous_split_chunks: bool = False):
    return split_tensor_along_dim(tensor, tensor.dim() - 1, num_partitions,
                                  contiguous_split_chunks)


def create_local_weight(full_weight: torch.Tensor, partition_dim: int,
                        per_partition_size: int, stride: int,
                        rank: Optional[int] = None,
                        world_size: Optional[int] = None,
                        out_weight: Optional[torch.Tensor] = None):
    """Slice this rank's shard out of a full (master) weight.

    Reference: layers.py:87-106 ``create_local_weight`` — with stride>1 the
    weight is a concatenation of ``stride`` sub-blocks each sharded
    separately (fused gate-up / QKV layouts).
    """
    from . import parallel_state as ps

    if rank is None:
        rank = ps.get_tensor_model_parallel_rank()
    if world_size is None:
        world_size = ps.get_tensor_model_parallel_size()

    per_stride_size = divide(per_partition_size, stride)
    chunks = torch.split(full_weight, per_stride_size, dim=partition_dim)
    # chunks laid out as [world*stride] sub-blocks; rank r takes blocks
    # r, r+world, r+2*world, ... (one per stride block)
    my = [chunks[rank + world_size * s] for s in range(stride)]
    cat = torch.cat(my, dim=partition_dim)
    if out_weight is not None:
        out_weight.data.copy_(cat)
        return out_weight
    return cat.contiguous()


def cast_if_autocast_enabled(*args):
    if not torch.is_autocast_enabled("cuda"):
        return args
    dtype = torch.get_autocast_dtype("cuda")

    def _cast(x):
        if isinstance(x, torch.Tensor) and x.is_floating_point():
            return x.to(dtype)
        return x

    return tuple(_cast(a) for a in args)


def move_all_tensor_to_cpu(obj: Any):
    if isinstance(obj, torch.Tensor):
        return obj.cpu()
    if isinstance(obj, dict):
        return {k: move_all_tensor_to_cpu(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        t = [move_all_tensor_to_cpu(v) for v in obj]
        return type(obj)(t) if not isinstance(obj, tuple) else tuple(t)
    return obj


def verify_casted_dtype(*tensors):
    dtypes = {t.dtype for t in tensors if isinstance(t, torch.Tensor)}
    if len(dtypes) > 1:
        raise RuntimeError(f"mixed dtypes after autocast: {dtypes}")
