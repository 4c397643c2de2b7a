"""Channel-parallel Conv2d layers (reference parallel_layers/layers.py:
1159-1540 ``OutputChannelParallelConv2d`` / ``InputChannelParallelConv2d``).

Output-channel sharding mirrors ColumnParallel (weight (O/tp, I, kH, kW));
input-channel sharding mirrors RowParallel (weight (O, I/tp, kH, kW), output
all-reduced).  Conv itself runs on MIOpen via F.conv2d."""


import torch
import torch.nn as nn
import torch.nn.functional as F

from . import parallel_state as ps
from .mappings import (
    copy_to_tensor_model_parallel_region,
    gather_from_tensor_model_parallel_region,
    reduce_from_tensor_model_parallel_region,
)
from .utils import divide, set_tensor_model_parallel_attributes


class _ParallelConv2dBase(nn.Module):
    def _init_weight(self, full_out, full_in, kernel_size, partition_dim,
                     init_method, dtype):
        if self.weight.device.type == "meta":
            return
        master = torch.empty(full_out, full_in, *kernel_size,
                             dtype=torch.float32, device="cpu")
        init_method(master)
        tp = ps.get_tensor_model_parallel_size()
        r = ps.get_tensor_model_parallel_rank()
        shard = master.chunk(tp, dim=partition_dim)[r]
        with torch.no_grad():
            self.weight.data.copy_(shard.to(self.weight.dtype))


class OutputChannelParallelConv2d(_ParallelConv2dBase):
    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0, bias=True, gather_output=True, dtype=None,
                 device=None, init_method=None):
        super().__init__()
        tp = ps.get_tensor_model_parallel_size()
        if isinstance(kernel_size, int):
            kernel_size = (kernel_size, kernel_size)
        dtype = dtype or torch.get_default_dtype()
        self.stride, self.padding = stride, padding
        self.gather_output = gather_output
        self.out_channels_per_partition = divide(out_channels, tp)
        self.weight = nn.Parameter(torch.empty(
            self.out_channels_per_partition, in_channels, *kernel_size,
            dtype=dtype, device=device))
        set_tensor_model_parallel_attributes(self.weight, tp > 1, 0, 1, tp)
        init_method = init_method or (
            lambda t: nn.init.kaiming_uniform_(t, a=5 ** 0.5))
        self._init_weight(out_channels, in_channels, kernel_size, 0,
                          init_method, dtype)
        if bias:
            self.bias = nn.Parameter(torch.zeros(
                self.out_channels_per_partition, dtype=dtype, device=device))
            set_tensor_model_parallel_attributes(self.bias, tp > 1, 0, 1, tp)
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        x = copy_to_tensor_model_parallel_region(x)
        out = F.conv2d(x, self.weight, self.bias, stride=self.stride,
                       padding=self.padding)
        if self.gather_output:
            # channels dim = 1
            out = out.movedim(1, -1)
            out = gather_from_tensor_model_parallel_region(out)
            out = out.movedim(-1, 1)
        return out


class InputChannelParallelConv2d(_ParallelConv2dBase):
    def __init__(self, in_channels, out_channels, kernel_size, stride=1,
                 padding=0, bias=True, input_is_parallel=True, dtype=None,
                 device=None, init_method=None):
        super().__init__()
        tp = ps.get_tensor_model_parallel_size()
        if isinstance(kernel_size, int):
            kernel_size = (kernel_size, kernel_size)
        dtype = dtype or torch.get_default_dtype()
        self.stride, self.padding = stride, padding
        self.input_is_parallel = input_is_parallel
        self.in_channels_per_partition = divide(in_channels, tp)
        self.weight = nn.Parameter(torch.empty(
            out_channels, self.in_channels_per_partition, *kernel_size,
            dtype=dtype, device=device))
        set_tensor_model_parallel_attributes(self.weight, tp > 1, 1, 1, tp)
        init_method = init_method or (
            lambda t: nn.init.kaiming_uniform_(t, a=5 ** 0.5))
        self._init_weight(out_channels, in_channels, kernel_size, 1,
                          init_method, dtype)
        if bias:
            self.bias = nn.Parameter(torch.zeros(out_channels, dtype=dtype,
                                                 device=device))
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        if not self.input_is_parallel:
            r = ps.get_tensor_model_parallel_rank()
            tp = ps.get_tensor_model_parallel_size()
            x = x.chunk(tp, dim=1)[r]
        out = F.conv2d(x, self.weight, None, stride=self.stride,
                       padding=self.padding)
        out = reduce_from_tensor_model_parallel_region(out)
        if self.bias is not None:
            out = out + self.bias.view(1, -1, 1, 1)
        return out
