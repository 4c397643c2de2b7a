"""Checkpoint engine (reference trainer/checkpoint.py:110-973).

Implements the reference's public behavior:
* per-rank shard files ``{tag}/model/dp_rank_xx_tp_rank_xx_pp_rank_xx.pt``
  (reference _get_path :54-63),
* optimizer saved per DP rank when ZeRO-1 (:760-787),
* async saves on a single-thread executor with a "done"-tag commit
  protocol + num_kept garbage collection (:66-98,110-324),
* ``load_checkpoint(tag='latest_if_exists')`` resume-on-restart and
  broadcast-free per-rank loads.
"""

import os
import shutil
from concurrent.futures import ThreadPoolExecutor
from typing import Any, Dict, Optional

import torch

from ..parallel import comm, parallel_state as ps
from ..utils.logger import get_logger

logger = get_logger(__name__)

_EXECUTOR: Optional[ThreadPoolExecutor] = None
_PENDING = []  # (future, path, tag, num_kept)


def _executor() -> ThreadPoolExecutor:
    global _EXECUTOR
    if _EXECUTOR is None:
        _EXECUTOR = ThreadPoolExecutor(max_workers=1)
        import atexit

        atexit.register(finalize_checkpoints)
    return _EXECUTOR


def finalize_checkpoints():
    """Flush pending async saves and commit their done-tags; runs on the
    MAIN thread (collectives are not thread-safe with RCCL) — called at the
    start of the next save and at exit (reference atexit flush :730-735)."""
    global _PENDING
    rank = ps._cur_rank() if ps.model_parallel_is_initialized() else 0
    for fut, path, tag, num_kept in _PENDING:
        fut.result()
        comm.barrier()
        if rank == 0:
            _write_done(path, tag)
            _gc_checkpoints(path, num_kept)
    _PENDING = []


def _ep_size() -> int:
    return ps._GROUPS["ep"].size if "ep" in ps._GROUPS else 1


def _cp_size() -> int:
    return ps._GROUPS["cp"].size if "cp" in ps._GROUPS else 1


def _rank_name(dp: bool = True) -> str:
    """Shard filename stem (reference _get_path, trainer/checkpoint.py:54-63):
    ``dp_rank_xx[_cp_rank_xx][_ep_rank_xx]_tp_rank_xx_pp_rank_xx``.
    The cp/ep segments appear only when that parallelism is on, so the
    tp/pp-only layout stays byte-identical to round-1 checkpoints.
    ``dp=False`` pins the dp (and cp) fields to 0 — used for model shards,
    which are deduplicated over data-parallel replicas."""
    name = f"dp_rank_{ps.get_data_parallel_rank() if dp else 0:02d}"
    if _cp_size() > 1:
        name += f"_cp_rank_{ps.get_context_model_parallel_rank() if dp else 0:02d}"
    if _ep_size() > 1:
        name += f"_ep_rank_{ps.get_expert_model_parallel_rank():02d}"
    name += (f"_tp_rank_{ps.get_tensor_model_parallel_rank():02d}"
             f"_pp_rank_{ps.get_pipeline_model_parallel_rank():02d}")
    return name


def _is_model_writer() -> bool:
    """Model shards are deduped over data-parallel replicas.  Without EP
    that is dp/cp rank 0.  With EP, each EP rank holds DIFFERENT experts,
    so one writer per EP rank: the rank with expert-data-parallel rank 0
    (the EDP dim spans DPxCP)."""
    if _ep_size() > 1:
        return ps.get_expert_data_parallel_rank() == 0
    return (ps.get_data_parallel_rank() == 0
            and ps.get_context_model_parallel_rank() == 0)


def _done_file(path: str, tag: str) -> str:
    return os.path.join(path, str(tag), "done")


def _write_done(path: str, tag: str):
    with open(_done_file(path, tag), "w") as f:
        f.write("done")


def checkpoint_exists(path: str, tag: str) -> bool:
    return os.path.exists(_done_file(path, tag))


def has_checkpoint(path: str) -> bool:
    return len(_list_checkpoints(path)) > 0


def _list_checkpoints(path: str):
    if not os.path.isdir(path):
        return []
    tags = []
    for t in os.listdir(path):
        if os.path.exists(_done_file(path, t)):
            tags.append(t)

    def key(t):
        try:
            return (0, int(t))
        except ValueError:
            return (1, t)

    return sorted(tags, key=key)


def _gc_checkpoints(path: str, num_kept: Optional[int]):
    """Delete oldest completed checkpoints beyond num_kept, and any
    interrupted (done-less) ones (reference :66-98,259-272)."""
    if not os.path.isdir(path):
        return
    complete = _list_checkpoints(path)
    for t in os.listdir(path):
        full = os.path.join(path, t)
        if os.path.isdir(full) and t not in complete and \
                any(x in os.listdir(full) for x in ("model", "optim")):
            logger.warning("removing interrupted checkpoint %s", full)
            shutil.rmtree(full, ignore_errors=True)
    if num_kept is not None and len(complete) > num_kept:
        for t in complete[:-num_kept]:
            shutil.rmtree(os.path.join(path, t), ignore_errors=True)


def _save_obj(obj: Any, fname: str):
    os.makedirs(os.path.dirname(fname), exist_ok=True)
    cpu_obj = _to_cpu(obj)
    torch.save(cpu_obj, fname)


def _to_cpu(obj):
    if isinstance(obj, torch.Tensor):
        return obj.detach().cpu()
    if isinstance(obj, dict):
        return {k: _to_cpu(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        t = [_to_cpu(v) for v in obj]
        return t if isinstance(obj, list) else tuple(t)
    return obj


def save_checkpoint(path: str, tag, model=None, optimizer=None,
                    scheduler=None, user_content: Optional[Dict] = None,
                    num_workers: int = 8, use_xser: bool = False,  # noqa: ARG001 — xser is an XLA host-RAM workaround; MI355X snapshots to CPU and torch.saves (API compat)
                    num_kept: Optional[int] = None, async_save: bool = False,
                    zero1_optimizer: Optional[bool] = None) -> None:
    """reference trainer/checkpoint.py:654-824."""
    tag = str(tag)
    ckpt_dir = os.path.join(path, tag)
    rank = ps._cur_rank() if ps.model_parallel_is_initialized() else 0
    os.makedirs(ckpt_dir, exist_ok=True)

    jobs = []
    if model is not None:
        sd = model.state_dict()
        if _is_model_writer():
            jobs.append((sd, os.path.join(ckpt_dir, "model",
                                          _rank_name(dp=False) + ".pt")))
    if optimizer is not None:
        from ..optimizer import NeuronZero1Optimizer
        from .optimizer import NxDOptimizer

        inner = optimizer.optimizer if isinstance(optimizer, NxDOptimizer) \
            else optimizer
        is_zero1 = isinstance(inner, NeuronZero1Optimizer) \
            if zero1_optimizer is None else zero1_optimizer
        # zero1: every DP rank holds distinct shards -> all ranks write
        if is_zero1 or ps.get_data_parallel_rank() == 0:
            jobs.append((optimizer.state_dict(),
                         os.path.join(ckpt_dir, "optim",
                                      _rank_name() + ".pt")))
    if scheduler is not None and rank == 0:
        jobs.append((scheduler.state_dict(),
                     os.path.join(ckpt_dir, "scheduler.pt")))
    if user_content is not None and rank == 0:
        jobs.append((user_content, os.path.join(ckpt_dir, "user_content.pt")))

    def commit():
        for obj, fname in jobs:
            _save_obj(obj, fname)
        comm.barrier()
        if rank == 0:
            _write_done(path, tag)
            _gc_checkpoints(path, num_kept)
        comm.barrier()

    if async_save:
        finalize_checkpoints()  # drain previous async save first
        # snapshot tensors to CPU NOW, write files in background
        jobs = [(_to_cpu(o), f) for o, f in jobs]
        comm.barrier()

        def bg():
            for obj, fname in jobs:
                _save_obj(obj, fname)

        _PENDING.append((_executor().submit(bg), path, tag, num_kept))
    else:
        commit()


def load_checkpoint(path: str, tag=None, model=None, optimizer=None,
                    scheduler=None, strict: bool = True):
    """reference trainer/checkpoint.py:347-432; tag=None -> latest complete.
    Returns user_content (or None)."""
    if tag is None or tag == "latest_if_exists":
        tags = _list_checkpoints(path)
        if not tags:
            if tag == "latest_if_exists":
                return None
            raise FileNotFoundError(f"no complete checkpoint under {path}")
        tag = tags[-1]
    tag = str(tag)
    ckpt_dir = os.path.join(path, tag)
    if not checkpoint_exists(path, tag):
        raise FileNotFoundError(f"checkpoint {ckpt_dir} incomplete (no done tag)")

    if model is not None:
        # model shards are written deduped with dp/cp pinned to 0
        fname = os.path.join(ckpt_dir, "model", _rank_name(dp=False) + ".pt")
        sd = torch.load(fname, map_location="cpu", weights_only=False)
        model.load_state_dict(sd, strict=strict)
    if optimizer is not None:
        fname = os.path.join(ckpt_dir, "optim", _rank_name() + ".pt")
        if not os.path.exists(fname):
            # non-zero1 optimizers are saved by dp rank 0 only
            fname = os.path.join(ckpt_dir, "optim",
                                 _rank_name(dp=False) + ".pt")
        optimizer.load_state_dict(
            torch.load(fname, map_location="cpu", weights_only=False))
    if scheduler is not None:
        f = os.path.join(ckpt_dir, "scheduler.pt")
        if os.path.exists(f):
            scheduler.load_state_dict(
                torch.load(f, map_location="cpu", weights_only=False))
    uc = os.path.join(ckpt_dir, "user_content.pt")
    if os.path.exists(uc):
        return torch.load(uc, map_location="cpu", weights_only=False)
    return None
