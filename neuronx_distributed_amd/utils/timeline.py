"""Chrome-trace timeline (reference utils/timeline.py:15-141 ``Timeline``):
mark_event_start/end per label; per-step gather of every rank's events over
gloo; rank 0 dumps chrome://tracing JSON."""

import json
import os
import time
from typing import Optional

import torch.distributed as dist

from ..parallel import parallel_state as ps


class Timeline:
    def __init__(self, trace_file_path: Optional[str], rank: Optional[int] = None):
        self.enabled = trace_file_path is not None
        self.path = trace_file_path
        self.rank = rank if rank is not None else (
            dist.get_rank() if dist.is_initialized() else 0)
        self.events = []
        self._open = {}

    def mark_event_start(self, label: str):
        if self.enabled:
            self._open[label] = time.perf_counter_ns() // 1000

    def mark_event_end(self, label: str):
        if not self.enabled or label not in self._open:
            return
        t0 = self._open.pop(label)
        t1 = time.perf_counter_ns() // 1000
        self.events.append({"name": label, "ph": "X", "ts": t0,
                            "dur": t1 - t0, "pid": self.rank, "tid": 0})

    def mark_step_end(self, gather: bool = True):
        """Gather all ranks' events and dump from rank 0 (reference
        :92-126 + parallel_state.py:1581-1590 gather_python_object)."""
        if not self.enabled:
            return
        all_events = [self.events]
        if gather and dist.is_initialized() and dist.get_world_size() > 1:
            gathered = [None] * dist.get_world_size()
            dist.all_gather_object(gathered, self.events)
            all_events = gathered
        if self.rank == 0:
            flat = [e for evs in all_events for e in evs]
            os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
            with open(self.path, "w") as f:
                json.dump({"traceEvents": flat}, f)
        self.events = []


class PPTimeline(Timeline):
    """Pipeline-task instrumentation (reference pipeline/timeline.py)."""

    def mark_task(self, task_name: str, mb: int):
        return _TaskCtx(self, f"{task_name}_mb{mb}")


class _TaskCtx:
    def __init__(self, tl, label):
        self.tl = tl
        self.label = label

    def __enter__(self):
        self.tl.mark_event_start(self.label)

    def __exit__(self, *a):
        self.tl.mark_event_end(self.label)
        return False
