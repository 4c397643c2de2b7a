"""Autograd collective region transitions for TP / SP / EP.

API parity with the reference's ``parallel_layers/mappings.py`` (678 LoC;
exports at mappings.py:406-556): ``copy_to/reduce_from/scatter_to/
gather_from_tensor_model_parallel_region``, the ``*_sequence_parallel_region``
family, and ``enter/exit_expert_parallel_region`` — re-implemented as plain
eager autograd.Functions over RCCL (no XLA graph semantics needed).
"""

import torch

from . import comm
from . import parallel_state as ps


def _tp_group():
    return ps.get_group_info("tp")


def _split_along_dim(tensor, dim, group):
    world = comm.group_size(group)
    if world == 1:
        return tensor
    rank = comm.group_rank(group)
    assert tensor.shape[dim] % world == 0, (
        f"dim {dim} size {tensor.shape[dim]} not divisible by tp {world}"
    )
    return tensor.chunk(world, dim=dim)[rank].contiguous()


# ---------------------------------------------------------------------------
# TP region (reference mappings.py:176-279)
# ---------------------------------------------------------------------------

class _CopyToModelParallelRegion(torch.autograd.Function):
    """Identity forward; all-reduce grad backward (mappings.py:190-193)."""

    @staticmethod
    def forward(ctx, input_):
        return input_

    @staticmethod
    def backward(ctx, grad_output):
        grad = grad_output.contiguous()
        comm.all_reduce(grad, group=_tp_group())
        return grad


class _ReduceFromModelParallelRegion(torch.autograd.Function):
    """All-reduce forward; identity backward (mappings.py:43-52)."""

    @staticmethod
    def forward(ctx, input_):
        out = input_.contiguous()
        comm.all_reduce(out, group=_tp_group())
        return out

    @staticmethod
    def backward(ctx, grad_output):
        return grad_output


class _ScatterToModelParallelRegion(torch.autograd.Function):
    """Split last dim forward; all-gather backward."""

    @staticmethod
    def forward(ctx, input_):
        return _split_along_dim(input_, -1, _tp_group())

    @staticmethod
    def backward(ctx, grad_output):
        return comm.all_gather(grad_output, dim=grad_output.dim() - 1,
                               group=_tp_group())


class _GatherFromModelParallelRegion(torch.autograd.Function):
    """All-gather last dim forward; split backward (mappings.py:84-103)."""

    @staticmethod
    def forward(ctx, input_):
        return comm.all_gather(input_, dim=input_.dim() - 1, group=_tp_group())

    @staticmethod
    def backward(ctx, grad_output):
        return _split_along_dim(grad_output, -1, _tp_group())


def copy_to_tensor_model_parallel_region(input_):
    return _CopyToModelParallelRegion.apply(input_)


def reduce_from_tensor_model_parallel_region(input_):
    return _ReduceFromModelParallelRegion.apply(input_)


def scatter_to_tensor_model_parallel_region(input_):
    return _ScatterToModelParallelRegion.apply(input_)


def gather_from_tensor_model_parallel_region(input_):
    return _GatherFromModelParallelRegion.apply(input_)


# ---------------------------------------------------------------------------
# SP region (reference mappings.py:256-352): activations sharded on the
# sequence dim (dim 0 in (S,B,H) layout — we use seq_dim argument, default 0)
# ---------------------------------------------------------------------------

class _ScatterToSequenceParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_, seq_dim):
        ctx.seq_dim = seq_dim
        return _split_along_dim(input_, seq_dim, _tp_group())

    @staticmethod
    def backward(ctx, grad_output):
        return comm.all_gather(grad_output, dim=ctx.seq_dim, group=_tp_group()), None


class _GatherFromSequenceParallelRegion(torch.autograd.Function):
    """Fwd all-gather along seq; bwd reduce-scatter (when the consumer is a
    TP-region GEMM, to_model_parallel=True) or plain split."""

    @staticmethod
    def forward(ctx, input_, seq_dim, to_model_parallel):
        ctx.seq_dim = seq_dim
        ctx.to_model_parallel = to_model_parallel
        return comm.all_gather(input_, dim=seq_dim, group=_tp_group())

    @staticmethod
    def backward(ctx, grad_output):
        if ctx.to_model_parallel:
            g = comm.reduce_scatter(grad_output, dim=ctx.seq_dim, group=_tp_group())
        else:
            g = _split_along_dim(grad_output, ctx.seq_dim, _tp_group())
        return g, None, None


class _ReduceScatterToSequenceParallelRegion(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_, seq_dim):
        ctx.seq_dim = seq_dim
        return comm.reduce_scatter(input_, dim=seq_dim, group=_tp_group())

    @staticmethod
    def backward(ctx, grad_output):
        return comm.all_gather(grad_output, dim=ctx.seq_dim, group=_tp_group()), None


def scatter_to_sequence_parallel_region(input_, seq_dim=0):
    return _ScatterToSequenceParallelRegion.apply(input_, seq_dim)


def gather_from_sequence_parallel_region(input_, seq_dim=0, to_model_parallel=True):
    return _GatherFromSequenceParallelRegion.apply(input_, seq_dim, to_model_parallel)


def reduce_scatter_to_sequence_parallel_region(input_, seq_dim=0):
    return _ReduceScatterToSequenceParallelRegion.apply(input_, seq_dim)


# ---------------------------------------------------------------------------
# EP region (reference mappings.py:160-172,481-556): all-to-all swapping the
# expert dim against the token/capacity dim.
# ---------------------------------------------------------------------------

class _AllToAll(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input_, split_dim, concat_dim, group_name):
        ctx.split_dim = split_dim
        ctx.concat_dim = concat_dim
        ctx.group_name = group_name
        return comm.all_to_all(input_, split_dim, concat_dim,
                               group=ps.get_group_info(group_name))

    @staticmethod
    def backward(ctx, grad_output):
        g = comm.all_to_all(grad_output.contiguous(), ctx.concat_dim, ctx.split_dim,
                            group=ps.get_group_info(ctx.group_name))
        return g, None, None, None


def all_to_all(input_, split_dim, concat_dim, group_name="ep"):
    return _AllToAll.apply(input_, split_dim, concat_dim, group_name)


class _GatherFromGroup(torch.autograd.Function):
    """All-gather along ``dim`` over an arbitrary group; backward
    reduce-scatters the gradient (each rank's input chunk receives grad
    contributions from every rank's use of the gathered tensor)."""

    @staticmethod
    def forward(ctx, input_, dim, group_name):
        ctx.dim = dim
        ctx.group_name = group_name
        return comm.all_gather(input_, dim=dim,
                               group=ps.get_group_info(group_name))

    @staticmethod
    def backward(ctx, grad_output):
        g = comm.reduce_scatter(grad_output.contiguous(), dim=ctx.dim,
                                group=ps.get_group_info(ctx.group_name))
        return g, None, None


class _ReduceScatterToGroup(torch.autograd.Function):
    """Reduce-scatter along ``dim``; backward all-gathers the gradient."""

    @staticmethod
    def forward(ctx, input_, dim, group_name):
        ctx.dim = dim
        ctx.group_name = group_name
        return comm.reduce_scatter(input_.contiguous(), dim=dim,
                                   group=ps.get_group_info(group_name))

    @staticmethod
    def backward(ctx, grad_output):
        g = comm.all_gather(grad_output.contiguous(), dim=ctx.dim,
                            group=ps.get_group_info(ctx.group_name))
        return g, None, None


def gather_from_group(input_, dim=0, group_name="ep"):
    return _GatherFromGroup.apply(input_, dim, group_name)


def reduce_scatter_to_group(input_, dim=0, group_name="ep"):
    return _ReduceScatterToGroup.apply(input_, dim, group_name)


def enter_expert_parallel_region(hidden: torch.Tensor, scatter_gather: bool = False):
    """(E, C, H) -> (E/ep, ep*C, H): each EP rank ends up holding the full
    token set for its local experts (reference mappings.py:481-523)."""
    return _AllToAll.apply(hidden, 0, 1, "ep")


def exit_expert_parallel_region(hidden: torch.Tensor, scatter_gather: bool = False):
    """(E/ep, ep*C, H) -> (E, C, H): inverse of enter (mappings.py:525-556)."""
    return _AllToAll.apply(hidden, 1, 0, "ep")
