from .model_builder import ModelBuilder
from .nxd_model import NxDModel, StateInitializer
from .parallel_context import NxDParallelState
from .functions import (
    trace,
    compile,
    compile_wlo,
    compile_layout_transformer,
    shard_checkpoint,
    load_sharded_checkpoint,
)
from .kv_cache import (KVCache, RollingKVCache,
                       build_kv_caches)
from .generation import generate
