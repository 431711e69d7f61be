from .wrapper import PipelineParallelWrapper, distribute_layers
from .schedule import (
    PipelineSchedule,
    AllFwdAllBwdSchedule,
    OneFOneBSchedule,
    get_schedule,
)
from .trainer import PipelineTrainer
from .dataloader import PipelineDataLoader

__all__ = [
    "PipelineParallelWrapper",
    "distribute_layers",
    "PipelineSchedule",
    "AllFwdAllBwdSchedule",
    "OneFOneBSchedule",
    "get_schedule",
    "PipelineTrainer",
    "PipelineDataLoader",
]
