"""Quantization configuration (reference quantization/quantization_config.py:
25-35,100-125).

CDNA4-native dtypes: INT8, FP8 OCP ``e4m3fn`` and ``e5m2`` (gfx950 uses the
OCP formats, NOT MI300X's fnuz — cdna_hip_programming.md §4).  The
reference's ±240 E4M3 clamp is a Neuron-hardware range limit and does not
carry over."""

import enum
from dataclasses import dataclass

import torch


class QuantizedDtype(enum.Enum):
    INT8 = "int8"
    F8E4M3 = "f8e4m3fn"   # OCP e4m3fn (gfx950-native MFMA input)
    F8E5M2 = "f8e5m2"
    # packed storage formats (reference quantization_config.py:100-125):
    # 4 fp8 lanes per uint32 / 4 fp4 nibbles per uint16.  CDNA4's fp8 MFMA
    # consumes the unpacked view (a zero-copy byte reinterpret for fp8).
    F8E4M3FN_X4 = "f8e4m3fn_x4"
    F8E5M2_X4 = "f8e5m2_x4"
    F4E2M1FN_X4 = "f4e2m1fn_x4"

    @property
    def packed_count(self) -> int:
        return 4 if self in (QuantizedDtype.F8E4M3FN_X4,
                             QuantizedDtype.F8E5M2_X4,
                             QuantizedDtype.F4E2M1FN_X4) else 1

    @property
    def unpacked(self) -> "QuantizedDtype":
        """Element dtype a packed format stores."""
        return {
            QuantizedDtype.F8E4M3FN_X4: QuantizedDtype.F8E4M3,
            QuantizedDtype.F8E5M2_X4: QuantizedDtype.F8E5M2,
        }.get(self, self)

    @property
    def torch_dtype(self):
        return {
            QuantizedDtype.INT8: torch.int8,
            QuantizedDtype.F8E4M3: torch.float8_e4m3fn,
            QuantizedDtype.F8E5M2: torch.float8_e5m2,
            QuantizedDtype.F8E4M3FN_X4: torch.uint32,
            QuantizedDtype.F8E5M2_X4: torch.uint32,
            QuantizedDtype.F4E2M1FN_X4: torch.uint16,
        }[self]

    @property
    def max_value(self):
        return {
            QuantizedDtype.INT8: 127.0,
            QuantizedDtype.F8E4M3: 448.0,
            QuantizedDtype.F8E5M2: 57344.0,
            QuantizedDtype.F8E4M3FN_X4: 448.0,
            QuantizedDtype.F8E5M2_X4: 57344.0,
            QuantizedDtype.F4E2M1FN_X4: 6.0,
        }[self]


class QuantizationType(enum.Enum):
    PER_TENSOR_SYMMETRIC = "per_tensor_symmetric"
    PER_CHANNEL_SYMMETRIC = "per_channel_symmetric"


@dataclass
class QuantizationConfig:
    quantized_dtype: QuantizedDtype = QuantizedDtype.INT8
    quantization_type: QuantizationType = QuantizationType.PER_CHANNEL_SYMMETRIC
    quantization_per_channel_axis: int = 0
    # W8A8: dynamically quantize activations per-token to fp8 and run the
    # GEMM on the gfx950 fp8 MFMA pipe via torch._scaled_mm (~2.1x the
    # bf16 GEMM rate measured on MI355X).  False = weight-only (W8A16),
    # dequant->bf16 GEMM.
    quantize_activation: bool = False
