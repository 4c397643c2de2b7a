from .model import NxDPPModel
from .manual_pipe_stage import PipelineStageModule
from .scheduler import (
    InferenceSchedule,
    Train1F1BSchedule,
    TrainInterleavedSchedule,
)
from . import comm, partition, scheduler
