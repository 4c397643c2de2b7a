"""MoE module utilities (reference modules/moe/model_utils.py:1-110):
activation registry, GLU type/activation enums, default block size."""

import enum
from typing import Callable, Dict

import torch
import torch.nn.functional as F

from ..utils.logger import get_logger

logger = get_logger(__name__)

DEFAULT_BLOCK_SIZE = 512  # reference model_utils.py:94

ACT2FN: Dict[str, Callable] = {
    "gelu": F.gelu,
    "leaky_relu": F.leaky_relu,
    "relu": F.relu,
    "sigmoid": torch.sigmoid,
    "silu": F.silu,
    "tanh": torch.tanh,
}


class GLUType(enum.Enum):
    """Supported gated-linear-unit flavors (reference :20-36)."""

    GLU = "glu"
    SWIGLU = "swiglu"

    @classmethod
    def validate(cls, glu_type):
        if isinstance(glu_type, cls):
            return glu_type
        if glu_type is None:
            logger.warning("glu_type is None, defaulting to basic GLU")
            glu_type = "glu"
        if glu_type not in [e.value for e in cls]:
            raise ValueError(
                f"glu_type={glu_type} not supported, must be one of "
                f"{[e.value for e in cls]}")
        return cls(glu_type)


class ACTFunc(enum.Enum):
    """Activation ids matched with the HIP kernel enum values
    (reference :38-60)."""

    def __new__(cls, idx, name):
        obj = object.__new__(cls)
        obj._value_ = idx
        obj.id = idx
        obj.name_str = name
        return obj

    SILU = (0, "silu")
    GELU = (1, "gelu")
    GELU_TANH_APPROX = (2, "gelu_tanh_approx")
    SIGMOID = (3, "sigmoid")
    RELU = (4, "relu")
    TANH = (5, "tanh")
    LEAKY_RELU = (6, "leaky_relu")

    @classmethod
    def from_name(cls, name: str) -> "ACTFunc":
        for e in cls:
            if e.name_str == name:
                return e
        raise ValueError(f"unsupported activation {name}")
