from .wrapper import InterleavedPipelineWrapper, PipelineParallelWrapper, distribute_layers
from .schedule import (
    InterleavedOneFOneBSchedule,
    PipelineSchedule,
    AllFwdAllBwdSchedule,
    OneFOneBSchedule,
    get_schedule,
)
from .trainer import PipelineTrainer
from .dataloader import PipelineDataLoader

__all__ = [
    "InterleavedPipelineWrapper",
    "InterleavedOneFOneBSchedule",
    "PipelineParallelWrapper",
    "distribute_layers",
    "PipelineSchedule",
    "AllFwdAllBwdSchedule",
    "OneFOneBSchedule",
    "get_schedule",
    "PipelineTrainer",
    "PipelineDataLoader",
]
