"""Device-memory accounting helpers (reference exposes Neuron runtime
memory via NEURON_RT env tooling; on MI355X torch.cuda's allocator stats
are the native equivalent — 288 GB HBM3E per GPU)."""

from typing import Dict

import torch

from .logger import get_logger

logger = get_logger(__name__)


def memory_stats(device=None) -> Dict[str, float]:
    """Allocated/reserved/peak in GiB (zeros on CPU)."""
    if not torch.cuda.is_available():
        return {"allocated_gib": 0.0, "reserved_gib": 0.0,
                "peak_allocated_gib": 0.0, "total_gib": 0.0}
    gib = 1 << 30
    free, total = torch.cuda.mem_get_info(device)
    return {
        "allocated_gib": torch.cuda.memory_allocated(device) / gib,
        "reserved_gib": torch.cuda.memory_reserved(device) / gib,
        "peak_allocated_gib": torch.cuda.max_memory_allocated(device) / gib,
        "free_gib": free / gib,
        "total_gib": total / gib,
    }


def log_memory_stats(tag: str = "", device=None) -> Dict[str, float]:
    s = memory_stats(device)
    logger.info("[mem%s] alloc %.1f GiB | reserved %.1f | peak %.1f | "
                "free %.1f / %.1f GiB",
                f" {tag}" if tag else "", s["allocated_gib"],
                s["reserved_gib"], s["peak_allocated_gib"],
                s.get("free_gib", 0.0), s["total_gib"])
    return s
