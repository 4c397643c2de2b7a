"""ZeRO-1 optimizer, written from scratch for RCCL over xGMI.

Role parity with the reference's ``optimizer/zero_redundancy_optimizer.py``
(which subclasses torch-xla's ZeroRedundancyOptimizer, :9-30, and the EP
variant :163-381).  The reference delegates the heavy lifting to torch-xla;
here the whole machinery is ours:

* Parameters are packed into flat *buckets* (default cap 130 MB like the
  reference's reduce-scatter cap, trainer/trainer.py:258-280; env
  ``ALL_GATHER_REDUCE_SCATTER_BUCKET_CAP_MB`` overrides).  The model
  parameters are re-pointed into the flat bf16 buffer (zero-copy
  all-gather target) and their ``.grad`` views into the flat grad buffer
  (autograd accumulates straight into the reduce-scatter source).
* step(): per bucket — grad reduce-scatter over the sharding group
  (merged DPxCP; expert params shard over the EDP group instead,
  reference :241-281) → fp32 master-shard AdamW → param all-gather.
  All collectives are issued async and overlapped across buckets.
* Grad clipping matches ``get_grad_norm`` semantics (grads.py:41-189):
  TP-duplicated params counted once, norm all-reduced over sharding +
  TP + PP groups.

On GPU, the fp32 master AdamW update + bf16 write-back runs as one fused
HIP kernel per bucket shard (ops.adamw) instead of torch.optim's chain of
elementwise kernels.
"""

import math
import os
from typing import Dict, List

import torch
import torch.distributed as dist

from ..parallel import comm, parallel_state as ps
from ..parallel.utils import param_is_tensor_parallel, param_is_expert_parallel
from ..utils.logger import get_logger

logger = get_logger(__name__)

_DEFAULT_BUCKET_CAP_MB = 130  # reference trainer/trainer.py:258-280


def _bucket_cap_bytes():
    return int(os.environ.get("ALL_GATHER_REDUCE_SCATTER_BUCKET_CAP_MB",
                              _DEFAULT_BUCKET_CAP_MB)) * 1024 * 1024


class _Bucket:
    """One flat bucket: params become views of ``flat_param``; grads views
    of ``flat_grad``; this rank owns fp32 master shard ``master``."""

    def __init__(self, params: List[torch.nn.Parameter], group_info,
                 group_index: int, optimizer_dtype,
                 fp32_grad_acc: bool = False):
        self.params = params
        self.group_info = group_info
        self.group_index = group_index
        self.fp32_grad_acc = fp32_grad_acc
        world = group_info.size
        device = params[0].device
        dtype = params[0].dtype

        numel = sum(p.numel() for p in params)
        self.padded = int(math.ceil(numel / world) * world)
        self.flat_param = torch.zeros(self.padded, dtype=dtype, device=device)
        # fp32 grad accumulation (reference mixed_precision_config
        # use_fp32_grad_acc): grads accumulate into an fp32 flat buffer via
        # post-accumulate hooks (p.grad stays param-dtype per microbatch
        # and is folded in + freed); comm then runs in fp32
        grad_dtype = torch.float32 if fp32_grad_acc else dtype
        self.flat_grad = torch.zeros(self.padded, dtype=grad_dtype,
                                     device=device)

        # segment bookkeeping: (param, start, end)
        self.segments = []
        off = 0
        with torch.no_grad():
            for p in params:
                n = p.numel()
                self.flat_param[off:off + n].copy_(p.data.reshape(-1))
                new_data = self.flat_param[off:off + n].view(p.shape)
                p.data = new_data
                if fp32_grad_acc:
                    seg = self.flat_grad[off:off + n]

                    def _acc(param, _seg=seg):
                        if param.grad is not None:
                            _seg.add_(param.grad.detach().reshape(-1).float())
                            param.grad = None

                    p.register_post_accumulate_grad_hook(_acc)
                else:
                    p.grad = self.flat_grad[off:off + n].view(p.shape)
                self.segments.append((p, off, off + n))
                off += n

        self.shard_size = self.padded // world
        rank = group_info.rank_in_group(
            dist.get_rank() if dist.is_initialized() else 0)
        self.rank = rank
        lo = rank * self.shard_size
        hi = lo + self.shard_size
        self.shard_lo, self.shard_hi = lo, hi
        self.master = self.flat_param[lo:hi].to(optimizer_dtype)
        self.master.requires_grad_(True)
        # reduce-scatter lands in the comm dtype (= param dtype); the fp32
        # cast happens when attaching to the master shard
        self.grad_shard = torch.zeros(self.shard_size, dtype=grad_dtype,
                                      device=device)

        # which elements of MY shard must be excluded from the grad norm
        # (TP-duplicated params counted only on tp_rank 0)
        tp_rank = ps.get_tensor_model_parallel_rank()
        self.norm_exclude_ranges = []
        if tp_rank != 0:
            for p, s, e in self.segments:
                if not param_is_tensor_parallel(p):
                    s2, e2 = max(s, lo), min(e, hi)
                    if s2 < e2:
                        self.norm_exclude_ranges.append((s2 - lo, e2 - lo))
        # padding tail is zeros -> harmless for norm

    def relink_grads(self):
        """Re-point p.grad into the flat buffer if something (zero_grad
        set_to_none, checkpoint load) broke the linkage.  In fp32-acc mode
        fold in any grads the hooks have not consumed."""
        if self.fp32_grad_acc:
            for p, s, e in self.segments:
                if p.grad is not None:
                    self.flat_grad[s:e].add_(
                        p.grad.detach().reshape(-1).float())
                    p.grad = None
            return
        for p, s, e in self.segments:
            g = p.grad
            view = self.flat_grad[s:e].view(p.shape)
            if g is None:
                p.grad = view
            elif g.data_ptr() != view.data_ptr():
                view.add_(g)
                p.grad = view

    def local_sq_norm(self) -> torch.Tensor:
        # vector_norm with fp32 accumulation: ONE fused reduction, no
        # materialized fp32 copy of the (multi-GB) grad buffer
        sq = torch.linalg.vector_norm(self.grad_shard,
                                      dtype=torch.float32).pow(2)
        for s, e in self.norm_exclude_ranges:
            sq -= torch.linalg.vector_norm(self.grad_shard[s:e],
                                           dtype=torch.float32).pow(2)
        return sq


class NeuronZero1Optimizer(torch.optim.Optimizer):
    """ZeRO-1: optimizer states + master weights sharded over DP(xCP)."""

    def __init__(self, params, optimizer_class=torch.optim.AdamW,
                 optimizer_dtype=torch.float32, grad_clipping: bool = True,
                 max_norm: float = 1.0, pin_layout: bool = False,
                 sharding_groups=None, grad_norm_groups=None,
                 lazy_init: bool = False, use_fused_kernel: bool = True,
                 use_fp32_grad_acc: bool = False,
                 **defaults):
        if isinstance(params, torch.Tensor):
            raise TypeError("params must be an iterable")
        param_groups = list(params)
        if not param_groups:
            raise ValueError("empty parameter list")
        if not isinstance(param_groups[0], dict):
            param_groups = [{"params": param_groups}]

        self.optimizer_class = optimizer_class
        self.optimizer_dtype = optimizer_dtype
        self.grad_clipping = grad_clipping
        self.max_norm = max_norm
        self.use_fused_kernel = use_fused_kernel
        self.use_fp32_grad_acc = use_fp32_grad_acc
        self._sharding_group = (sharding_groups
                                if isinstance(sharding_groups, ps.GroupInfo)
                                else None)
        if self._sharding_group is None:
            self._sharding_group = ps.get_group_info("zero1") \
                if "zero1" in ps._GROUPS else ps.get_group_info("dp")

        super().__init__(param_groups, defaults)

        self.buckets: List[_Bucket] = []
        self._build_buckets()

        # base optimizer over master shards, preserving per-group options
        base_groups = []
        self._base_group_index = []  # base group -> source param_group idx
        for gi, group in enumerate(self.param_groups):
            opts = {k: v for k, v in group.items() if k != "params"}
            shard_params = [b.master for b in self.buckets if b.group_index == gi]
            if shard_params:
                base_groups.append({"params": shard_params, **opts})
                self._base_group_index.append(gi)
        self.base_optimizer = optimizer_class(base_groups, **{})
        self._grad_norm = None

    # -- construction -----------------------------------------------------

    def _shard_group_for(self, param):
        if param_is_expert_parallel(param) and "edp" in ps._GROUPS:
            return ps.get_group_info("edp")
        return self._sharding_group

    def _build_buckets(self):
        cap = _bucket_cap_bytes()
        for gi, group in enumerate(self.param_groups):
            # split params by sharding group (dense vs expert)
            by_shard: Dict[int, List] = {}
            for p in group["params"]:
                if not p.requires_grad:
                    continue
                key = id(self._shard_group_for(p))
                by_shard.setdefault(key, (self._shard_group_for(p), []))[1].append(p)
            for _, (ginfo, plist) in by_shard.items():
                cur, size = [], 0
                for p in plist:
                    cur.append(p)
                    size += p.numel() * p.element_size()
                    if size >= cap:
                        self.buckets.append(_Bucket(
                            cur, ginfo, gi, self.optimizer_dtype,
                            self.use_fp32_grad_acc))
                        cur, size = [], 0
                if cur:
                    self.buckets.append(_Bucket(
                        cur, ginfo, gi, self.optimizer_dtype,
                        self.use_fp32_grad_acc))

    # -- step -------------------------------------------------------------

    def _bucket_grad_scale(self, b) -> float:
        """Mean scale for one bucket's grads.  Dense buckets: 1/shard-group
        size.  Expert buckets (sharded over EDP): the grads of an expert
        already sum token contributions from all EP ranks (all-to-all
        backward), so they need an extra 1/ep on top of the 1/edp reduce
        scale to match the dense 1/(data-parallel) mean (reference
        zero_redundancy_optimizer.py:241-281 ep scale_factor)."""
        ep_size = ps._GROUPS["ep"].size if "ep" in ps._GROUPS else 1
        if ep_size > 1 and b.group_info.name == "edp":
            return 1.0 / (b.group_info.size * ep_size)
        return 1.0 / b.group_info.size

    @torch.no_grad()
    def _reduce_scatter_grads(self):
        works = []
        for b in self.buckets:
            world = b.group_info.size
            scale = self._bucket_grad_scale(b)
            if world == 1:
                # single-rank shard group: the shard IS the flat grad
                # (the 1/ep expert scale still applies when EP > 1)
                if scale != 1.0:
                    b.flat_grad.mul_(scale)
                b.grad_shard = b.flat_grad
                continue
            b.flat_grad.mul_(scale)  # grad mean over the shard group
            if comm._backend_is_gloo(b.group_info.group):
                dist.all_reduce(b.flat_grad, group=b.group_info.group)
                b.grad_shard.copy_(b.flat_grad[b.shard_lo:b.shard_hi])
            else:
                w = dist.reduce_scatter_tensor(b.grad_shard, b.flat_grad,
                                               group=b.group_info.group,
                                               async_op=True)
                works.append(w)
        for w in works:
            w.wait()

    @torch.no_grad()
    def _clip_grads(self, apply: bool = True):
        device = self.buckets[0].master.device
        sq = torch.zeros(1, dtype=torch.float32, device=device)
        for b in self.buckets:
            sq += b.local_sq_norm()
        # reduce over sharding group + TP + PP (reference grads.py:41-189)
        comm.all_reduce(sq, group=self._sharding_group)
        for name in ("tp", "pp"):
            if name in ps._GROUPS and ps._GROUPS[name].size > 1:
                comm.all_reduce(sq, group=ps._GROUPS[name])
        total_norm = sq.sqrt()
        self._grad_norm = total_norm.squeeze()
        clip = torch.clamp(self.max_norm / (total_norm + 1e-6), max=1.0)
        if apply:
            for b in self.buckets:
                b.grad_shard.mul_(clip)
        return clip

    @torch.no_grad()
    def _all_gather_params(self, copy_master: bool = True):
        works = []
        for b in self.buckets:
            shard = b.flat_param[b.shard_lo:b.shard_hi]
            if copy_master:
                shard.copy_(b.master.to(b.flat_param.dtype))
            world = b.group_info.size
            if world == 1:
                continue
            if comm._backend_is_gloo(b.group_info.group):
                parts = list(b.flat_param.chunk(world))
                dist.all_gather(parts, shard.clone(), group=b.group_info.group)
            else:
                w = dist.all_gather_into_tensor(b.flat_param, shard.clone(),
                                                group=b.group_info.group,
                                                async_op=True)
                works.append(w)
        for w in works:
            w.wait()

    def _use_fused(self) -> bool:
        from .. import ops

        return (self.use_fused_kernel
                and not self.use_fp32_grad_acc
                and self.optimizer_class is torch.optim.AdamW
                and self.optimizer_dtype == torch.float32
                and self.buckets and self.buckets[0].master.is_cuda
                and ops.is_available())

    @torch.no_grad()
    def _fused_step(self, clip):
        """One HIP kernel per bucket shard: bf16 grad -> fp32 m/v/master ->
        bf16 param written straight into the all-gather source slice."""
        from .. import ops

        self._step_count = getattr(self, "_step_count", 0) + 1
        for b in self.buckets:
            group = self.param_groups[b.group_index]
            if not hasattr(b, "fused_m"):
                b.fused_m = torch.zeros_like(b.master)
                b.fused_v = torch.zeros_like(b.master)
            beta1, beta2 = group.get("betas", (0.9, 0.999))
            ops.adamw_step(b.master, b.fused_m, b.fused_v, b.grad_shard,
                           b.flat_param[b.shard_lo:b.shard_hi], clip,
                           group.get("lr", 1e-3), beta1, beta2,
                           group.get("eps", 1e-8),
                           group.get("weight_decay", 0.0), self._step_count)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for b in self.buckets:
            b.relink_grads()
        self._reduce_scatter_grads()

        if self._use_fused():
            clip = None
            if self.grad_clipping:
                clip = self._clip_grads(apply=False)
            self._fused_step(clip)
            self._all_gather_params(copy_master=False)
            return loss

        if self.grad_clipping:
            self._clip_grads()
        # LR schedulers mutate OUR param_groups; mirror hyperparams onto
        # the base optimizer's (copied) groups before stepping
        for bg, gi in zip(self.base_optimizer.param_groups,
                          self._base_group_index):
            for k, v in self.param_groups[gi].items():
                if k != "params":
                    bg[k] = v
        for b in self.buckets:
            b.master.grad = b.grad_shard.to(self.optimizer_dtype)
        self.base_optimizer.step()
        for b in self.buckets:
            b.master.grad = None
        self._all_gather_params()
        return loss

    @property
    def grad_norm(self):
        return self._grad_norm

    def zero_grad(self, set_to_none: bool = False):
        for b in self.buckets:
            b.flat_grad.zero_()
            if b.fp32_grad_acc:
                for p, _, _ in b.segments:
                    p.grad = None
                continue
            for p, s, e in b.segments:
                if p.grad is None or p.grad.data_ptr() != b.flat_grad[s:e].data_ptr():
                    p.grad = b.flat_grad[s:e].view(p.shape)

    # -- checkpoint -------------------------------------------------------

    def state_dict(self, include_masters: bool = True):
        """``include_masters=False`` (mixed_precision_config
        use_master_weights_in_ckpt=False) drops the fp32 master shards —
        smaller checkpoints; on load they are rebuilt from the bf16
        params (one-time precision round-trip)."""
        return {
            "base_optimizer": self.base_optimizer.state_dict(),
            "step_count": getattr(self, "_step_count", 0),
            "fused_state": [
                {"m": b.fused_m.cpu(), "v": b.fused_v.cpu()}
                if hasattr(b, "fused_m") else None for b in self.buckets
            ],
            "masters": [b.master.detach().cpu() for b in self.buckets]
            if include_masters else None,
            "shard_meta": [
                {"padded": b.padded, "rank": b.rank,
                 "world": b.group_info.size, "group": b.group_info.name,
                 "group_index": b.group_index,
                 # per-param segment map so the OFFLINE dp-shard merge
                 # (scripts/convert_zero_checkpoints.py, reference
                 # optimizer/convert_zero_checkpoints.py:15-179) can
                 # reconstruct full per-parameter tensors
                 "segments": [(s, e, tuple(p.shape))
                              for p, s, e in b.segments]}
                for b in self.buckets
            ],
        }

    def load_state_dict(self, state_dict):
        self.base_optimizer.load_state_dict(state_dict["base_optimizer"])
        self._step_count = state_dict.get("step_count", 0)
        for b, fs in zip(self.buckets, state_dict.get("fused_state", [])):
            if fs is not None:
                b.fused_m = fs["m"].to(b.master.device)
                b.fused_v = fs["v"].to(b.master.device)
        masters = state_dict.get("masters")
        if masters is None:
            # masters were excluded from the checkpoint: rebuild from the
            # (already-loaded) bf16 params
            for b in self.buckets:
                b.master.data.copy_(
                    b.flat_param[b.shard_lo:b.shard_hi].to(b.master.dtype))
        else:
            for b, m in zip(self.buckets, masters):
                b.master.data.copy_(m.to(b.master.device))
        self._all_gather_params()


class NeuronEPZero1Optimizer(NeuronZero1Optimizer):
    """EP variant (reference :163-381): expert params shard over the EDP
    group — handled generically by ``_shard_group_for``; this subclass
    exists for API parity."""
    pass
