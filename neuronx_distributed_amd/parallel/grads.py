"""Gradient synchronization + clipping across the parallel dimensions.

Parity with reference ``parallel_layers/grads.py`` (367 LoC):
``bucket_allreduce_gradients`` (:259-327, 512 MB buckets, reverse order),
``clip_grad_norm``/``get_grad_norm`` (:41-256),
``allreduce_sequence_parallel_gradients`` (:330-346) and
``allreduce_context_parallel_gradients`` (:348-366).

MI355X: buckets are sized for xGMI ring throughput; the all-reduce runs on
RCCL and can overlap backward when driven from grad hooks (trainer).
"""

import os
from typing import Iterable, List, Optional

import torch
import torch.distributed as dist

from . import comm
from . import parallel_state as ps
from .utils import param_is_tensor_parallel, param_is_expert_parallel

# 512 MB default like the reference (grads.py:34); xGMI rings sustain peak
# well below this, override with ALLREDUCE_BUCKET_CAP_MB.
_DEFAULT_BUCKET_CAP_MB = 512


def _bucket_cap_bytes() -> int:
    return float(os.environ.get("ALLREDUCE_BUCKET_CAP_MB", _DEFAULT_BUCKET_CAP_MB)) * 1024 * 1024


def _flat_allreduce(grads: List[torch.Tensor], group, scale: float = 1.0):
    flat = torch._utils._flatten_dense_tensors(grads)
    if scale != 1.0:
        flat.mul_(scale)
    comm.all_reduce(flat, group=group)
    for g, synced in zip(grads, torch._utils._unflatten_dense_tensors(flat, grads)):
        g.copy_(synced)


def bucket_allreduce_gradients(grads: Iterable[torch.Tensor], group=None,
                               scale: Optional[float] = None):
    """Bucketed all-reduce, reverse order, bucketed per dtype (reference
    grads.py:259-327).  ``scale`` defaults to 1/group_size (grad mean)."""
    if group is None:
        group = ps.get_group_info("dp")
    world = comm.group_size(group)
    if world == 1:
        return
    if scale is None:
        scale = 1.0 / world

    grads = [g for g in grads if g is not None]
    cap = _bucket_cap_bytes()
    buckets = {}
    sizes = {}
    for g in reversed(list(grads)):  # reverse layer order for overlap parity
        key = g.dtype
        buckets.setdefault(key, []).append(g)
        sizes[key] = sizes.get(key, 0) + g.numel() * g.element_size()
        if sizes[key] >= cap:
            _flat_allreduce(buckets.pop(key), group, scale)
            sizes[key] = 0
    for key, bucket in buckets.items():
        if bucket:
            _flat_allreduce(bucket, group, scale)


def allreduce_gradients_for_parameters(parameters, group=None, scale=None):
    """DP gradient sync (non-ZeRO path), expert-aware (reference
    trainer/optimizer.py:132-141 + grads.py:284-298): dense grads all-reduce
    over the DP group; expert-parallel grads are distinct per EP rank, so
    they all-reduce over the expert-data-parallel (EDP) group only, with an
    extra 1/ep scale so dense and expert grads see the same effective
    1/(data-parallel-degree) mean."""
    parameters = [p for p in parameters if p.grad is not None]
    ep_size = (ps._GROUPS["ep"].size
               if "ep" in ps._GROUPS and group is None else 1)
    if ep_size > 1:
        expert = [p.grad for p in parameters if param_is_expert_parallel(p)]
        dense = [p.grad for p in parameters
                 if not param_is_expert_parallel(p)]
        if expert:
            edp = ps.get_group_info("edp")
            if edp.size == 1:
                # no replicas to reduce, but the 1/ep mean scale still applies
                for g in expert:
                    g.mul_(1.0 / ep_size)
            else:
                bucket_allreduce_gradients(
                    expert, group=edp, scale=1.0 / (edp.size * ep_size))
        bucket_allreduce_gradients(dense, group=group, scale=scale)
        return
    grads = [p.grad for p in parameters]
    bucket_allreduce_gradients(grads, group=group, scale=scale)


def allreduce_sequence_parallel_gradients(parameters):
    """All-reduce grads of params marked ``sequence_parallel_enabled`` over
    TP (layernorm weights replicated across TP in SP mode; reference
    grads.py:330-346)."""
    grads = [
        p.grad for p in parameters
        if p.grad is not None and getattr(p, "sequence_parallel_enabled", False)
    ]
    if grads:
        _flat_allreduce(grads, ps.get_group_info("tp"))


def allreduce_context_parallel_gradients(parameters):
    """CP duplicates non-attention compute: all-reduce all grads over CP and
    scale by 1/cp (reference grads.py:348-366)."""
    if ps.get_context_model_parallel_size() == 1:
        return
    grads = [p.grad for p in parameters if p.grad is not None]
    if grads:
        cp = ps.get_group_info("cp")
        _flat_allreduce(grads, cp, scale=1.0 / cp.size)


def get_grad_norm(parameters, norm_type: float = 2.0,
                  zero1_optimizer: bool = False, zero1_groups=None) -> torch.Tensor:
    """Global grad norm across TP/PP/EP (+DP sharding groups under ZeRO-1),
    reference grads.py:41-189.

    TP-duplicated (non-parallel) params are only counted on tp_rank 0 so the
    cross-rank reduction doesn't multi-count them.
    """
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    params_with_grad = [p for p in parameters if p.grad is not None]
    tp_rank = ps.get_tensor_model_parallel_rank()
    ep_size = ps._GROUPS["ep"].size if "ep" in ps._GROUPS else 1

    device = None
    for p in params_with_grad:
        device = p.grad.device
        break
    if device is None:
        device = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    # Expert params are DISTINCT per EP rank (sum their norm over EP);
    # dense params are REPLICATED across EP (count once per EP group) —
    # so the two accumulate separately (reference ep_total_norm,
    # grads.py:41-189).
    dense_acc = torch.zeros(1, dtype=torch.float32, device=device)
    expert_acc = torch.zeros(1, dtype=torch.float32, device=device)
    use_inf = norm_type == float("inf")
    for p in params_with_grad:
        is_tp = param_is_tensor_parallel(p)
        if not is_tp and tp_rank != 0:
            continue  # TP-duplicated: count once
        acc = expert_acc if param_is_expert_parallel(p) else dense_acc
        g = p.grad.detach()
        if use_inf:
            torch.maximum(acc, g.abs().max().reshape(1).float(), out=acc)
        elif norm_type == 2.0:
            # fused single-pass reduction with fp32 accumulation
            acc += torch.linalg.vector_norm(g, dtype=torch.float32).pow(2)
        else:
            acc += g.detach().abs().float().pow(norm_type).sum()

    # Reduce over every model-parallel dim; ZeRO-1 also reduces over the DP
    # sharding group since each rank only holds a shard's grads.  Expert
    # norms additionally reduce over EP (before joining the dense part).
    shared_groups = []
    for name in ("tp", "pp"):
        if name in ps._GROUPS and ps._GROUPS[name].size > 1:
            shared_groups.append(ps._GROUPS[name])
    if zero1_optimizer:
        if zero1_groups is not None:
            shared_groups.append(zero1_groups)
        elif "dp" in ps._GROUPS and ps._GROUPS["dp"].size > 1:
            shared_groups.append(ps._GROUPS["dp"])

    op = dist.ReduceOp.MAX if use_inf else dist.ReduceOp.SUM
    # Under ZeRO-1 every rank's (dense + expert) shard pieces are mutually
    # distinct across the whole dp*cp sharding group, so the sharding-group
    # reduce already covers EP; reducing over EP again would count each
    # expert piece ep times.
    if ep_size > 1 and not zero1_optimizer:
        comm.all_reduce(expert_acc, op=op, group=ps._GROUPS["ep"])
    if use_inf:
        local = torch.maximum(dense_acc, expert_acc)
        for g in shared_groups:
            comm.all_reduce(local, op=dist.ReduceOp.MAX, group=g)
        return local.squeeze()
    local = dense_acc + expert_acc
    for g in shared_groups:
        comm.all_reduce(local, group=g)
    return local.squeeze().pow(1.0 / norm_type)


def clip_grad_norm(parameters, max_norm: float, norm_type: float = 2.0,
                   zero1_optimizer: bool = False, zero1_groups=None) -> torch.Tensor:
    """Clip to ``max_norm``; returns the pre-clip total norm (reference
    grads.py:192-256)."""
    if isinstance(parameters, torch.Tensor):
        parameters = [parameters]
    parameters = list(parameters)
    total_norm = get_grad_norm(parameters, norm_type=norm_type,
                               zero1_optimizer=zero1_optimizer,
                               zero1_groups=zero1_groups)
    clip_coeff = max_norm / (total_norm + 1.0e-6)
    clip_coeff = torch.clamp(clip_coeff, max=1.0)
    for p in parameters:
        if p.grad is not None:
            p.grad.detach().mul_(clip_coeff.to(p.grad.device))
    return total_norm
