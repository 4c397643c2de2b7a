"""Lightning strategy (reference lightning/strategy.py:36
``NeuronXLAStrategy``): initializes torch.distributed + model parallelism
in setup_distributed, exposes the parallel ranks to the trainer."""

from typing import Optional

import torch

try:
    import lightning.pytorch as pl  # noqa: F401
    from lightning.pytorch.strategies import DDPStrategy as _Base

    _HAVE_LIGHTNING = True
except Exception:  # pragma: no cover - lightning not installed in image
    _Base = object
    _HAVE_LIGHTNING = False

from ..parallel import parallel_state as ps
from ..utils.logger import get_logger

logger = get_logger(__name__)


class NeuronLTStrategy(_Base):
    """MI355X strategy: one process per GPU over RCCL, model parallelism
    from an nxd_config (reference strategy.py:95-110 setup_distributed)."""

    def __init__(self, nxd_config=None, tensor_parallel_size: int = 1,
                 pipeline_parallel_size: int = 1, **kwargs):
        if not _HAVE_LIGHTNING:
            raise ImportError(
                "lightning is not installed; NeuronLTStrategy needs it")
        super().__init__(**kwargs)
        self.nxd_config = nxd_config
        self.tensor_parallel_size = (
            nxd_config["tensor_parallel_size"] if nxd_config
            else tensor_parallel_size)
        self.pipeline_parallel_size = (
            nxd_config["pipeline_parallel_size"] if nxd_config
            else pipeline_parallel_size)

    def setup_distributed(self):
        super().setup_distributed()
        if not ps.model_parallel_is_initialized():
            ps.initialize_model_parallel(
                tensor_model_parallel_size=self.tensor_parallel_size,
                pipeline_model_parallel_size=self.pipeline_parallel_size)

    @property
    def distributed_sampler_kwargs(self):
        return dict(num_replicas=ps.get_data_parallel_size(),
                    rank=ps.get_data_parallel_rank())
