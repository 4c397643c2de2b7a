"""Module-swap quantization (reference quantization/quantize.py:18-145
``convert``)."""

from typing import Optional

import torch.nn as nn

from ..parallel.layers import ColumnParallelLinear, RowParallelLinear
from ..utils.logger import get_logger
from .quantization_config import QuantizationConfig
from .quantization_layers import QuantizedColumnParallel, QuantizedRowParallel

logger = get_logger(__name__)

_MAPPING = {
    ColumnParallelLinear: QuantizedColumnParallel,
    RowParallelLinear: QuantizedRowParallel,
}


def convert(module: nn.Module, q_config: Optional[QuantizationConfig] = None,
            inplace: bool = True, mapping=None,
            include: Optional[list] = None) -> nn.Module:
    """Swap every Column/RowParallel linear for its quantized twin."""
    q_config = q_config or QuantizationConfig()
    mapping = mapping or _MAPPING
    n = 0
    for name, parent in list(module.named_modules()):
        for cname, child in list(parent.named_children()):
            if include is not None and cname not in include:
                continue
            qcls = mapping.get(type(child))
            if qcls is not None:
                setattr(parent, cname,
                        qcls.from_float(child, q_config))
                n += 1
    logger.info("quantized %d parallel linears (%s)", n,
                q_config.quantized_dtype.value)
    return module
