"""Training metrics utilities — parity with the reference example helpers
(reference examples/training/llama/training_utils.py:343-369 ``Throughput``
moving average and tp_zero1_llama_hf_pretrain.py:63-131 ``TrainingMetrics``
JSON writer), kept in the library so every example/integration script
shares them.
"""

import json
import os
import time
from collections import deque
from typing import Optional


class Throughput:
    """Moving-average throughput in sequences/second over a window of
    optimizer steps (reference training_utils.py:343-369)."""

    def __init__(self, batch_size: int, world_size: int,
                 grad_accum_usteps: int = 1, moving_avg_window_size: int = 10):
        self.seqs_per_iteration = batch_size * world_size * grad_accum_usteps
        self.moving_avg_window = deque(maxlen=moving_avg_window_size)
        self.start = time.time()

    def get_throughput(self) -> float:
        """Call once per optimizer step; returns seq/s averaged over the
        last ``moving_avg_window_size`` steps."""
        now = time.time()
        self.moving_avg_window.append(now - self.start)
        self.start = now
        return self.seqs_per_iteration * len(self.moving_avg_window) / \
            max(sum(self.moving_avg_window), 1e-9)


class TrainingMetrics:
    """Append run metrics to a JSON results file (reference
    tp_zero1_llama_hf_pretrain.py:63-131): a list of entries
    ``{"MetricName": ..., "MeasuredValue": ..., "Units": ...,
    "Timestamp": ...}`` plus run parameters."""

    def __init__(self, json_file: str):
        self.json_file = json_file

    def read_modify_write_file(self, data, key: str = "metrics") -> None:
        if os.path.exists(self.json_file):
            with open(self.json_file) as f:
                try:
                    result_dict = json.load(f) or {}
                except json.JSONDecodeError:
                    result_dict = {}
        else:
            result_dict = {}
        if isinstance(data, dict):
            result_dict.setdefault(key, {}).update(data)
        elif key in result_dict:
            result_dict[key].extend(data)
        else:
            result_dict[key] = list(data)
        with open(self.json_file, "w") as f:
            json.dump(result_dict, f, indent=2)

    def store_metrics(self, metrics) -> None:
        data = [{
            "MetricName": m.name,
            "MeasuredValue": m.value,
            "Units": m.units,
            "Timestamp": m.timestamp,
            "AdditionalData": m.additional_data,
        } for m in metrics]
        self.read_modify_write_file(data, key="metrics")

    def store_parameters(self, parameters: dict) -> None:
        self.read_modify_write_file(parameters, key="parameters")


class Metric:
    def __init__(self, name: str, value, units: str = "",
                 additional_data: Optional[dict] = None):
        self.name = name
        self.value = value
        self.units = units
        self.timestamp = time.strftime("%Y-%m-%dT%H:%M:%S")
        self.additional_data = additional_data or {}
