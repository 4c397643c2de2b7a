"""Tensor utilities (reference utils/tensor_utils.py:4-60).

The reference implements cumsum via blocked tril matmuls because XLA's
native cumsum lowered poorly on Neuron; on MI355X ``torch.cumsum`` maps to
a tuned ROCm scan kernel, so the API is kept (same signature, same
fp64-accumulation contract) on top of the native op."""

import torch


def cumsum(tensor: torch.Tensor, dim: int = 0,
           tril_size: int = 2048) -> torch.Tensor:  # noqa: ARG001
    """Cumulative sum along dim 0 of a 2-D tensor with fp64 accumulation
    (reference signature/semantics; ``tril_size`` is the reference's
    matmul-block knob and is irrelevant to the native scan)."""
    if tensor.dim() != 2:
        raise ValueError(f"Expected 2D input tensor, got {tuple(tensor.shape)}")
    if dim != 0:
        raise NotImplementedError("Only cumsum along dimension-0 is supported")
    return torch.cumsum(tensor.to(torch.float64), dim=0).to(tensor.dtype)
