"""Quantize/dequantize helpers (reference quantization/quantization_utils.py
+ dequantize.py:79)."""

import torch

from .quantization_config import (QuantizationConfig, QuantizationType,
                                  QuantizedDtype)


def quantize_symmetric(weight: torch.Tensor, cfg: QuantizationConfig):
    """Returns (q_weight, scale) with weight ~= q_weight * scale."""
    qmax = cfg.quantized_dtype.max_value
    if cfg.quantization_type == QuantizationType.PER_TENSOR_SYMMETRIC:
        amax = weight.abs().max().clamp(min=1e-8)
        scale = (amax / qmax).float()
    else:
        axis = cfg.quantization_per_channel_axis
        dims = [d for d in range(weight.dim()) if d != axis]
        amax = weight.abs().amax(dim=dims, keepdim=True).clamp(min=1e-8)
        scale = (amax / qmax).float()
    q = (weight.float() / scale)
    if cfg.quantized_dtype == QuantizedDtype.INT8:
        q = q.round().clamp(-qmax, qmax).to(torch.int8)
    else:
        q = q.clamp(-qmax, qmax).to(cfg.quantized_dtype.torch_dtype)
    return q, scale


def dequantize(q_weight: torch.Tensor, scale: torch.Tensor,
               dtype=torch.bfloat16) -> torch.Tensor:
    return (q_weight.float() * scale.float()).to(dtype)
