"""Quantize/dequantize helpers (reference quantization/quantization_utils.py
+ dequantize.py:79)."""

import torch

from .quantization_config import (QuantizationConfig, QuantizationType,
                                  QuantizedDtype)


# fp4 e2m1 value grid (sign x {0, .5, 1, 1.5, 2, 3, 4, 6}); nibble =
# sign<<3 | code
_FP4_GRID = torch.tensor([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0])


def _fp4_encode(x: torch.Tensor) -> torch.Tensor:
    """fp32 -> nearest e2m1 nibble code (uint8 in [0, 15])."""
    grid = _FP4_GRID.to(x.device)
    sign = (x < 0).to(torch.uint8)
    mag = x.abs().clamp(max=6.0)
    code = (mag.unsqueeze(-1) - grid).abs().argmin(dim=-1).to(torch.uint8)
    return (sign << 3) | code


def _fp4_decode(nib: torch.Tensor) -> torch.Tensor:
    grid = _FP4_GRID.to(nib.device)
    mag = grid[(nib & 0x7).long()]
    return torch.where((nib >> 3) > 0, -mag, mag)


def pack_x4(q: torch.Tensor, dtype: QuantizedDtype) -> torch.Tensor:
    """Pack 4 quantized lanes along the LAST dim (reference *_X4 storage
    formats, quantization_config.py:100-125): fp8 -> uint32 (byte
    reinterpret), fp4 nibbles -> uint16."""
    assert q.shape[-1] % 4 == 0, "last dim must be divisible by 4"
    if dtype == QuantizedDtype.F4E2M1FN_X4:
        nib = q.to(torch.int32).reshape(*q.shape[:-1], -1, 4)
        packed = (nib[..., 0] | (nib[..., 1] << 4) | (nib[..., 2] << 8)
                  | (nib[..., 3] << 12))
        return packed.to(torch.int32).to(torch.uint16)
    return q.contiguous().view(torch.uint8).reshape(
        *q.shape[:-1], -1, 4).contiguous().view(torch.uint32).squeeze(-1)


def unpack_x4(packed: torch.Tensor, dtype: QuantizedDtype) -> torch.Tensor:
    """Inverse of pack_x4; fp8 unpack is a zero-copy byte reinterpret."""
    if dtype == QuantizedDtype.F4E2M1FN_X4:
        p = packed.to(torch.int32)
        nibs = torch.stack([(p >> s) & 0xF for s in (0, 4, 8, 12)], dim=-1)
        return nibs.reshape(*packed.shape[:-1], -1).to(torch.uint8)
    out = packed.unsqueeze(-1).contiguous().view(torch.uint8)
    return out.reshape(*packed.shape[:-1], -1).view(
        dtype.unpacked.torch_dtype)


def quantize_symmetric(weight: torch.Tensor, cfg: QuantizationConfig):
    """Returns (q_weight, scale) with weight ~= q_weight * scale.  Packed
    *_X4 dtypes return the packed storage tensor."""
    dt = cfg.quantized_dtype
    qmax = dt.max_value
    if cfg.quantization_type == QuantizationType.PER_TENSOR_SYMMETRIC:
        amax = weight.abs().max().clamp(min=1e-8)
        scale = (amax / qmax).float()
    else:
        axis = cfg.quantization_per_channel_axis
        dims = [d for d in range(weight.dim()) if d != axis]
        amax = weight.abs().amax(dim=dims, keepdim=True).clamp(min=1e-8)
        scale = (amax / qmax).float()
    q = (weight.float() / scale)
    if dt == QuantizedDtype.INT8:
        q = q.round().clamp(-qmax, qmax).to(torch.int8)
    elif dt == QuantizedDtype.F4E2M1FN_X4:
        q = pack_x4(_fp4_encode(q.clamp(-qmax, qmax)), dt)
    elif dt.packed_count == 4:
        q = pack_x4(q.clamp(-qmax, qmax).to(dt.unpacked.torch_dtype), dt)
    else:
        q = q.clamp(-qmax, qmax).to(dt.torch_dtype)
    return q, scale


def dequantize(q_weight: torch.Tensor, scale: torch.Tensor,
               dtype=torch.bfloat16,
               quantized_dtype: QuantizedDtype = None) -> torch.Tensor:
    if quantized_dtype is not None and quantized_dtype.packed_count == 4:
        if quantized_dtype == QuantizedDtype.F4E2M1FN_X4:
            vals = _fp4_decode(unpack_x4(q_weight, quantized_dtype))
        else:
            vals = unpack_x4(q_weight, quantized_dtype).float()
        return (vals.float() * scale.float()).to(dtype)
    return (q_weight.float() * scale.float()).to(dtype)


def fp8_scaled_linear(x: torch.Tensor, q_weight: torch.Tensor,
                      w_scale: torch.Tensor,
                      out_dtype=torch.bfloat16) -> torch.Tensor:
    """W8A8 fp8 GEMM on the gfx950 fp8 MFMA pipe: dynamic per-token
    activation quantization (e4m3fn, rowwise scale) + per-channel weight
    scales through ``torch._scaled_mm`` (hipBLASLt fp8; ~2.1x the bf16
    rate on MI355X).  x (..., K) @ q_weight (N, K)^T -> (..., N)."""
    lead = x.shape[:-1]
    K = x.shape[-1]
    x2 = x.reshape(-1, K)
    amax = x2.abs().amax(dim=1, keepdim=True).float().clamp(min=1e-6)
    xs = amax / 448.0
    xq = (x2 / xs.to(x2.dtype)).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
    N = q_weight.shape[0]
    ws = w_scale.float().reshape(-1)
    if ws.numel() == 1:
        ws = ws.expand(N)
    sb = ws.reshape(1, N).contiguous()
    out = torch._scaled_mm(xq, q_weight.t(), scale_a=xs, scale_b=sb,
                           out_dtype=out_dtype)
    return out.reshape(*lead, N)
