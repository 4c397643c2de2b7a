"""OCP Microscaling (MX) quantization — torch emulation (reference
experimental/quantization/microscaling/mx_torch.py:65-203).

MXFP8/MXFP4: per-32-element blocks share an e8m0 power-of-two scale.  On
CDNA4 this maps natively onto ``v_mfma_scale_*_f8f6f4`` (the ONLY path to
the 5/10 PF low-precision peaks, cdna_hip_programming.md §3/§4); this
module provides the numerics-faithful emulation used for quantized-weight
preparation and accuracy studies."""

from typing import Tuple

import torch

MX_BLOCK = 32

_FMT = {
    "fp8_e4m3": (448.0, torch.float8_e4m3fn),
    "fp8_e5m2": (57344.0, torch.float8_e5m2),
    # fp4 e2m1: emulated via rounding to the 16-value grid
    "fp4_e2m1": (6.0, None),
}

_FP4_GRID = torch.tensor([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0])


def _round_fp4(x: torch.Tensor) -> torch.Tensor:
    sign = x.sign()
    mag = x.abs().clamp(max=6.0)
    grid = _FP4_GRID.to(x.device)
    idx = torch.bucketize(mag, (grid[1:] + grid[:-1]) / 2)
    return sign * grid[idx]


def quantize_mx(x: torch.Tensor, fmt: str = "fp8_e4m3", axis: int = -1
                ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Quantize along ``axis`` in blocks of 32 with shared e8m0 scales.
    Returns (q, scales) with x ~= q * 2**scales (q in the element format,
    kept as the emulation dtype)."""
    emax, qdtype = _FMT[fmt]
    x = x.movedim(axis, -1)
    orig = x.shape
    assert orig[-1] % MX_BLOCK == 0
    xb = x.reshape(*orig[:-1], orig[-1] // MX_BLOCK, MX_BLOCK).float()
    amax = xb.abs().amax(dim=-1, keepdim=True).clamp(min=2.0 ** -126)
    # e8m0: power-of-two scale so the block max lands INSIDE the element
    # range (ceil: scaled max in (emax/2, emax], never overflowing e4m3fn's
    # finite-only encoding)
    scales = torch.ceil(torch.log2(amax / emax)).clamp(-127, 127)
    scaled = xb / torch.exp2(scales)
    if fmt == "fp4_e2m1":
        q = _round_fp4(scaled)
    else:
        q = scaled.to(qdtype).float()
    q = q.reshape(orig).movedim(-1, axis)
    return q, scales.squeeze(-1)


def dequantize_mx(q: torch.Tensor, scales: torch.Tensor, axis: int = -1,
                  dtype=torch.bfloat16) -> torch.Tensor:
    x = q.movedim(axis, -1).float()
    orig = x.shape
    xb = x.reshape(*orig[:-1], orig[-1] // MX_BLOCK, MX_BLOCK)
    out = xb * torch.exp2(scales.float()).unsqueeze(-1)
    return out.reshape(orig).movedim(-1, axis).to(dtype)


def mx_matmul(a: torch.Tensor, qb: torch.Tensor, b_scales: torch.Tensor,
              dtype=torch.bfloat16) -> torch.Tensor:
    """a (bf16) @ dequant(qb) — emulation of the scaled-MFMA GEMM."""
    return a.to(dtype) @ dequantize_mx(qb, b_scales, axis=-2, dtype=dtype)
