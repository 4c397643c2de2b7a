"""Single-process SPMD build context (reference trace/parallel_context.py:8
``NxDParallelState`` + trace/mock_torchdist.py:8-84).

Inside the context, ``initialize_model_parallel`` installs rank-list-only
groups (``group=None``) for an arbitrary world size and collectives become
identities — letting one process build/shard rank-r models without any
process group."""

import contextlib

from ..parallel import parallel_state as ps


class NxDParallelState(contextlib.AbstractContextManager):
    def __init__(self, world_size: int, rank: int = 0,
                 tensor_model_parallel_size: int = None,
                 pipeline_model_parallel_size: int = 1,
                 expert_model_parallel_size: int = 1):
        self.world_size = world_size
        self.rank = rank
        self.tp = tensor_model_parallel_size or world_size
        self.pp = pipeline_model_parallel_size
        self.ep = expert_model_parallel_size
        self._was_initialized = False

    def __enter__(self):
        self._was_initialized = ps.model_parallel_is_initialized()
        self._saved_groups = dict(ps._GROUPS)
        ps.enter_aot_mode(self.world_size, self.rank)
        ps._GROUPS = {}
        for name in ("_TENSOR_MODEL_PARALLEL_GROUP", "_DATA_PARALLEL_GROUP"):
            setattr(ps, name, None)
        ps.initialize_model_parallel(
            tensor_model_parallel_size=self.tp,
            pipeline_model_parallel_size=self.pp,
            expert_model_parallel_size=self.ep)
        return self

    def __exit__(self, *exc):
        ps.exit_aot_mode()
        ps._GROUPS = self._saved_groups
        return False
