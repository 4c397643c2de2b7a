"""Expert-fused 3-D tensor-parallel linears (reference
modules/moe/moe_parallel_layers.py:18-377).

Weights carry a leading local-expert dim: ColumnParallel ``(E_local, H,
I/tp)`` (sharded dim 2), RowParallel ``(E_local, I/tp, H)`` (sharded dim 1).
Forward is a batched GEMM ``einsum('ech,ehi->eci')`` -> hipBLASLt bmm.
Params are tagged ``expert_model_parallel`` so ZeRO-1 shards them over the
EDP group (reference :140-162 re-tagging)."""

import torch
import torch.nn as nn

from ..parallel import parallel_state as ps
from ..parallel.utils import (
    divide,
    set_tensor_model_parallel_attributes,
    EXPERT_PARALLEL_ATTR,
)


class _ExpertFusedLinearBase(nn.Module):
    def _tag(self, param, partition_dim):
        world = ps.get_tensor_model_parallel_size()
        set_tensor_model_parallel_attributes(param, world > 1, partition_dim,
                                             1, world)
        if ps.get_expert_model_parallel_size() > 1:
            setattr(param, EXPERT_PARALLEL_ATTR, True)

    def reset_parameters(self):
        """Meta-materialization: re-run the deterministic per-expert init."""
        args = getattr(self, "_reset_args", None)
        if args is None or not hasattr(self, "_reset_init"):
            return
        out_size, in_size, pdim = args
        init_method, dtype, stride = self._reset_init
        self._init_expertwise(self.weight, out_size, in_size, pdim,
                              init_method, dtype, stride=stride)

    def _init_expertwise(self, weight, full_out, full_in, partition_dim,
                         init_method, dtype, stride=1):
        """Deterministic per-expert init: full (H_in, H_out) master per
        expert on CPU, sliced to the TP shard (reference :247-261).  With
        ``stride>1`` the out dim is a concat of ``stride`` sub-blocks
        (fused [gate; up]) each sharded separately, so the local shard is
        ``[gate_r | up_r]`` (reference stride-2 fused layout)."""
        if weight.device.type == "meta":
            return
        tp = ps.get_tensor_model_parallel_size()
        tp_rank = ps.get_tensor_model_parallel_rank()
        with torch.no_grad():
            for e in range(weight.shape[0]):
                master = torch.empty(full_in, full_out, dtype=torch.float32,
                                     device="cpu")
                init_method(master)
                if partition_dim == 2:  # column: split out dim
                    blocks = master.chunk(stride, dim=1)
                    shard = torch.cat(
                        [b.chunk(tp, dim=1)[tp_rank] for b in blocks], dim=1)
                else:  # row: split in dim
                    shard = master.chunk(tp, dim=0)[tp_rank]
                weight.data[e].copy_(shard.to(weight.dtype))


class ExpertFusedColumnParallelLinear(_ExpertFusedLinearBase):
    def __init__(self, num_experts_local, input_size, output_size,
                 dtype=None, device=None, init_method=None, stride=1):
        super().__init__()
        world = ps.get_tensor_model_parallel_size()
        dtype = dtype or torch.get_default_dtype()
        self.output_size_per_partition = divide(output_size, world)
        self.weight = nn.Parameter(torch.empty(
            num_experts_local, input_size, self.output_size_per_partition,
            dtype=dtype, device=device))
        self._tag(self.weight, 2)
        self.weight.partition_stride = stride
        init_method = init_method or (lambda t: nn.init.normal_(t, std=0.02))
        self._reset_args = (output_size, input_size, 2)
        self._reset_init = (init_method, dtype, stride)
        self._init_expertwise(self.weight, output_size, input_size, 2,
                              init_method, dtype, stride=stride)

    def forward(self, x, expert_indices=None):
        # x (E_local, C, H) -> (E_local, C, I/tp); grad_input all-reduced
        # over TP in backward by the caller's input copy (delayed reduce).
        # expert_indices (N,): compute only those experts' slices, x (N,C,H)
        # (reference moe_parallel_layers.py:263-276 selective loading)
        w = self.weight
        if expert_indices is not None:
            w = w.index_select(0, expert_indices.reshape(-1))
        return torch.bmm(x, w.to(x.dtype))


class ExpertFusedRowParallelLinear(_ExpertFusedLinearBase):
    def __init__(self, num_experts_local, input_size, output_size,
                 dtype=None, device=None, init_method=None,
                 reduce_output=False):
        super().__init__()
        world = ps.get_tensor_model_parallel_size()
        dtype = dtype or torch.get_default_dtype()
        self.input_size_per_partition = divide(input_size, world)
        self.reduce_output = reduce_output
        self.weight = nn.Parameter(torch.empty(
            num_experts_local, self.input_size_per_partition, output_size,
            dtype=dtype, device=device))
        self._tag(self.weight, 1)
        init_method = init_method or (lambda t: nn.init.normal_(t, std=0.02))
        self._reset_args = (output_size, input_size, 1)
        self._reset_init = (init_method, dtype, 1)
        self._init_expertwise(self.weight, output_size, input_size, 1,
                              init_method, dtype)

    def forward(self, x, expert_indices=None):
        w = self.weight
        if expert_indices is not None:
            w = w.index_select(0, expert_indices.reshape(-1))
        out = torch.bmm(x, w.to(x.dtype))
        if self.reduce_output:
            from ..parallel.mappings import reduce_from_tensor_model_parallel_region

            out = reduce_from_tensor_model_parallel_region(out)
        return out
