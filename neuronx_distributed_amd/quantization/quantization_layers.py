"""Quantized tensor-parallel linears (reference
quantization/quantization_layers.py:465,744): mirror Column/RowParallel
config exactly (same shapes / partition dims :591-603) with the weight kept
in the quantized dtype + scale parameters; the GEMM runs dequant->bf16
hipBLASLt (CDNA4 fp8 MFMA GEMM path is a later optimization — the layout
and scale plumbing here is the contract)."""


import torch
import torch.nn as nn

from ..parallel import parallel_state as ps
from ..parallel.layers import (
    linear_with_async_allreduce,
    BaseParallelLinear,
)
from ..parallel.mappings import (
    copy_to_tensor_model_parallel_region,
    gather_from_tensor_model_parallel_region,
    reduce_from_tensor_model_parallel_region,
    scatter_to_tensor_model_parallel_region,
)
from ..parallel.utils import set_tensor_model_parallel_attributes
from .quantization_config import (QuantizationConfig, QuantizationType,
                                  QuantizedDtype)
from .quantization_utils import (dequantize, fp8_scaled_linear,
                                 quantize_symmetric, unpack_x4)


class _QuantizedParallelLinearBase(BaseParallelLinear):
    def _use_fp8_mm(self, input_: torch.Tensor) -> bool:
        """fp8 MFMA path: W8A8 e4m3fn via torch._scaled_mm (inference —
        the quantized weight carries no grad).  SP still runs the dequant
        path (the all-gather is fused into the bf16 linear there)."""
        cfg = self.quantization_config
        return (cfg.quantize_activation
                and cfg.quantized_dtype in (QuantizedDtype.F8E4M3,
                                            QuantizedDtype.F8E4M3FN_X4)
                and input_.is_cuda
                and not self.sequence_parallel_enabled
                and not input_.requires_grad)

    def _fp8_weight(self):
        """fp8 view of the stored weight; packed X4 unpacks zero-copy."""
        dt = self.quantization_config.quantized_dtype
        return self.weight if dt.packed_count == 1 else \
            unpack_x4(self.weight, dt)

    def _make_scale(self, out_rows: int, cfg: QuantizationConfig, shard: bool):
        if cfg.quantization_type == QuantizationType.PER_TENSOR_SYMMETRIC:
            scale = nn.Parameter(torch.ones(1), requires_grad=False)
        else:
            scale = nn.Parameter(torch.ones(out_rows, 1), requires_grad=False)
            if shard:
                set_tensor_model_parallel_attributes(
                    scale, ps.get_tensor_model_parallel_size() > 1, 0)
        return scale

    @classmethod
    def from_float(cls, float_layer, quantization_config=None):
        """Convert an existing Column/RowParallel layer (reference
        quantize.convert per-module path)."""
        cfg = quantization_config or QuantizationConfig()
        layer = cls.__new__(cls)
        BaseParallelLinear.__init__(layer)
        layer.quantization_config = cfg
        q, s = quantize_symmetric(float_layer.weight.detach(), cfg)
        layer.weight = nn.Parameter(q, requires_grad=False)
        for attr in ("tensor_model_parallel", "partition_dim",
                     "partition_stride", "num_partitions"):
            if hasattr(float_layer.weight, attr):
                setattr(layer.weight, attr, getattr(float_layer.weight, attr))
        layer.scale = nn.Parameter(
            s if s.dim() else s.reshape(1), requires_grad=False)
        layer.bias = float_layer.bias
        layer._copy_cfg(float_layer)
        return layer


class QuantizedColumnParallel(_QuantizedParallelLinearBase):
    def _copy_cfg(self, fl):
        self.gather_output = fl.gather_output
        self.sequence_parallel_enabled = fl.sequence_parallel_enabled
        self.compute_dtype = fl.dtype

    def forward(self, input_):
        if self._use_fp8_mm(input_):
            out = fp8_scaled_linear(input_, self._fp8_weight(), self.scale,
                                    self.compute_dtype)
            if self.bias is not None:
                out = out + self.bias
            if self.gather_output:
                out = gather_from_tensor_model_parallel_region(out)
            return out
        w = dequantize(self.weight, self.scale, self.compute_dtype,
                       self.quantization_config.quantized_dtype)
        if not self.sequence_parallel_enabled and \
                ps.get_tensor_model_parallel_size() > 1:
            input_parallel = input_
            async_ar = True
        else:
            input_parallel = copy_to_tensor_model_parallel_region(input_)
            async_ar = False
        out = linear_with_async_allreduce(
            input_parallel, w, self.bias, async_grad_allreduce=async_ar,
            sequence_parallel_enabled=self.sequence_parallel_enabled)
        if self.gather_output:
            out = gather_from_tensor_model_parallel_region(out)
        return out


class QuantizedRowParallel(_QuantizedParallelLinearBase):
    def _copy_cfg(self, fl):
        self.input_is_parallel = fl.input_is_parallel
        self.sequence_parallel_enabled = fl.sequence_parallel_enabled
        self.compute_dtype = fl.dtype

    def forward(self, input_):
        if not self.input_is_parallel:
            input_ = scatter_to_tensor_model_parallel_region(input_)
        if self._use_fp8_mm(input_):
            out = fp8_scaled_linear(input_, self._fp8_weight(), self.scale,
                                    self.compute_dtype)
        else:
            w = dequantize(self.weight, self.scale, self.compute_dtype,
                           self.quantization_config.quantized_dtype)
            out = linear_with_async_allreduce(
                input_, w, None, async_grad_allreduce=False,
                sequence_parallel_enabled=False)
        if self.sequence_parallel_enabled:
            from ..parallel.mappings import (
                reduce_scatter_to_sequence_parallel_region)

            out = reduce_scatter_to_sequence_parallel_region(out, seq_dim=0)
        else:
            out = reduce_from_tensor_model_parallel_region(out)
        if self.bias is not None:
            out = out + self.bias
        return out


class QuantizedParallelLinearLayerStateDictAdaptor:
    """Adapts externally-quantized (HF-style) checkpoints to the layout the
    quantized parallel layers expect (reference quantization_layers.py:356
    QuantizedParallelLinearLayerStateDictAdaptor): handles both plain
    ``{prefix}weight`` entries and torch.ao ``_packed_params`` entries
    produced by dynamic int8 quantization."""

    @staticmethod
    def get_weight_from_state_dict(prefix: str, state_dict: dict):
        if prefix + "weight" in state_dict:
            return state_dict[prefix + "weight"]
        if prefix + "_packed_params.dtype" in state_dict:
            qw = state_dict[prefix + "_packed_params._packed_params"][0]
            return torch.int_repr(qw)
        raise RuntimeError(f"Cannot find {prefix}weight in the state_dict")

    @staticmethod
    def set_weight_to_state_dict(prefix: str, tensor, state_dict: dict):
        if prefix + "weight" in state_dict:
            state_dict[prefix + "weight"] = tensor
        elif prefix + "_packed_params.dtype" in state_dict:
            packed = list(state_dict[prefix + "_packed_params._packed_params"])
            packed[0] = tensor
            state_dict[prefix + "_packed_params._packed_params"] = \
                tuple(packed)
        else:
            raise RuntimeError(
                f"Cannot find {prefix}weight in the state_dict")

    @staticmethod
    def get_scale_from_state_dict(prefix: str, state_dict: dict):
        for key in ("weight_scale", "scale"):
            if prefix + key in state_dict:
                return state_dict[prefix + key]
        if prefix + "_packed_params.dtype" in state_dict:
            qw = state_dict[prefix + "_packed_params._packed_params"][0]
            if qw.qscheme() in (torch.per_tensor_affine,
                                torch.per_tensor_symmetric):
                return torch.tensor([qw.q_scale()], dtype=torch.float32)
            return qw.q_per_channel_scales().float()
        raise RuntimeError(f"Cannot find {prefix}weight_scale in state_dict")

    @staticmethod
    def get_bias_from_state_dict(prefix: str, state_dict: dict):
        if prefix + "bias" in state_dict:
            return state_dict[prefix + "bias"]
        if prefix + "_packed_params.dtype" in state_dict:
            return state_dict[prefix + "_packed_params._packed_params"][1]
        return None
