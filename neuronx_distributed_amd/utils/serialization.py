"""Tensor/object (de)serialization for the pipeline engine (reference
utils/serialization.py:73-140 ``SerializationManager``/``TensorMeta``)."""

from dataclasses import dataclass
from typing import Any, List, Tuple

import torch


@dataclass
class TensorMeta:
    tensor_index: int
    dtype: torch.dtype
    shape: Tuple[int, ...]
    requires_grad: bool
    device: Any = None


class SerializationManager:
    """Split an arbitrary (nested) python object into (skeleton, metas,
    tensors) and rebuild it — the tensors travel over RCCL, the skeleton
    over the object channel."""

    def serialize(self, obj: Any):
        tensors: List[torch.Tensor] = []
        metas: List[TensorMeta] = []

        def strip(o):
            if isinstance(o, torch.Tensor):
                idx = len(tensors)
                tensors.append(o)
                metas.append(TensorMeta(idx, o.dtype, tuple(o.shape),
                                        o.requires_grad))
                return ("__tensor__", idx)
            if isinstance(o, dict):
                return {k: strip(v) for k, v in o.items()}
            if isinstance(o, (list, tuple)):
                t = [strip(v) for v in o]
                return t if isinstance(o, list) else ("__tuple__", t)
            return o

        return strip(obj), metas, tensors

    def deserialize(self, skeleton: Any, tensors: List[torch.Tensor]):
        def rebuild(o):
            if isinstance(o, tuple) and len(o) == 2 and o[0] == "__tensor__":
                return tensors[o[1]]
            if isinstance(o, tuple) and len(o) == 2 and o[0] == "__tuple__":
                return tuple(rebuild(v) for v in o[1])
            if isinstance(o, dict):
                return {k: rebuild(v) for k, v in o.items()}
            if isinstance(o, list):
                return [rebuild(v) for v in o]
            return o

        return rebuild(skeleton)
