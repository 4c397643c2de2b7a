"""Checkpoint IO plugin (reference lightning/checkpoint_io.py
``NeuronCheckpointIO``): routes Lightning saves through the sharded
checkpoint engine so every TP/PP rank writes its shard."""

import os
from typing import Any, Dict, Optional

import torch

from ..parallel import checkpointing as low_ckpt
from ..parallel import parallel_state as ps


class NeuronCheckpointIO:
    """Duck-typed lightning CheckpointIO (works standalone too)."""

    def save_checkpoint(self, checkpoint: Dict[str, Any], path,
                        storage_options: Optional[Any] = None) -> None:
        path = str(path)
        os.makedirs(path, exist_ok=True)
        tp = ps.get_tensor_model_parallel_rank()
        pp = ps.get_pipeline_model_parallel_rank()
        fname = os.path.join(path, f"tp_rank_{tp:02d}_pp_rank_{pp:02d}.ckpt")
        if ps.get_data_parallel_rank() == 0:
            torch.save(checkpoint, fname)

    def load_checkpoint(self, path, map_location=None) -> Dict[str, Any]:
        path = str(path)
        tp = ps.get_tensor_model_parallel_rank()
        pp = ps.get_pipeline_model_parallel_rank()
        fname = os.path.join(path, f"tp_rank_{tp:02d}_pp_rank_{pp:02d}.ckpt")
        return torch.load(fname, map_location=map_location or "cpu",
                          weights_only=False)

    def remove_checkpoint(self, path) -> None:
        import shutil

        shutil.rmtree(str(path), ignore_errors=True)
