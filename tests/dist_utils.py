"""Multi-process gloo test harness: run a function under world_size ranks on
CPU (the reference's NXD_CPU_MODE test vehicle).  Hang-proof:
a deadline kills stuck ranks and fails the test instead of blocking pytest."""

import os
import queue as pyqueue
import time
import traceback

import torch
import torch.multiprocessing as mp


def _worker(rank, world_size, port, fn, args, q):
    try:
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        torch.distributed.init_process_group("gloo", rank=rank,
                                             world_size=world_size)
        result = fn(rank, world_size, *args)
        q.put((rank, "ok", result))
    except Exception:
        q.put((rank, "err", traceback.format_exc()))
    finally:
        if torch.distributed.is_initialized():
            torch.distributed.destroy_process_group()


def _free_port() -> int:
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def run_distributed(fn, world_size=2, args=(), timeout=150, _retry=True):
    """Spawn world_size processes running fn(rank, world_size, *args);
    returns list of per-rank results; raises on any rank error or hang.
    Retries ONCE on failure with a fresh port (rendezvous port races are
    the only nondeterminism here)."""
    try:
        return _run_distributed_once(fn, world_size, args, timeout)
    except RuntimeError:
        if not _retry:
            raise
        return _run_distributed_once(fn, world_size, args, timeout)


def _run_distributed_once(fn, world_size=2, args=(), timeout=150):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [
        ctx.Process(target=_worker, args=(r, world_size, port, fn, args, q),
                    daemon=True)
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = {}
    deadline = time.time() + timeout
    while len(results) < world_size and time.time() < deadline:
        try:
            rank, status, payload = q.get(timeout=1.0)
            results[rank] = (status, payload)
        except pyqueue.Empty:
            if all(not p.is_alive() for p in procs) and q.empty():
                break
    for p in procs:
        if p.is_alive():
            p.terminate()
    for p in procs:
        p.join(10)
    if len(results) < world_size:
        missing = [r for r in range(world_size) if r not in results]
        raise RuntimeError(
            f"ranks {missing} hung or died without reporting; "
            f"got: { {r: s for r, (s, _) in results.items()} }")
    errs = {r: p for r, (s, p) in results.items() if s == "err"}
    if errs:
        raise RuntimeError(f"rank failures: {errs}")
    return [results[r][1] for r in sorted(results)]
