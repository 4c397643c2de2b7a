"""Multi-process gloo test harness: run a function under world_size ranks on
CPU (the reference's NXD_CPU_MODE test vehicle)."""

import os
import pickle
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


def run_distributed(fn, world_size=2, args=(), timeout=180):
    """Spawn world_size processes running fn(rank, world_size, *args);
    returns list of per-rank results; raises on any rank error."""
    import random

    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = random.randint(29600, 39999)
    procs = [
        ctx.Process(target=_worker, args=(r, world_size, port, fn, args, q))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world_size):
        rank, status, payload = q.get()
        results[rank] = (status, payload)
    for p in procs:
        p.join(timeout)
        if p.is_alive():
            p.terminate()
    errs = {r: p for r, (s, p) in results.items() if s == "err"}
    if errs:
        raise RuntimeError(f"rank failures: {errs}")
    return [results[r][1] for r in sorted(results)]
