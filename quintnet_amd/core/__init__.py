"""Core runtime: mesh, process groups, config, comm primitives."""

from .mesh import MeshGenerator
from .process_groups import ProcessGroupManager, init_process_groups
from .config import load_config, merge_configs, ParallelismConfig, TrainingConfig
from .comm import (
    Send,
    Recv,
    All_Gather,
    All_Reduce,
    ReduceScatter,
    copy_to_group,
    scatter_to_sequence,
    pipeline_communicate,
    bidirectional_pipeline_communicate,
)
from .distributed import (
    get_rank,
    get_world_size,
    get_local_rank,
    is_main_process,
    setup_distributed,
    cleanup_distributed,
    barrier,
)

__all__ = [
    "MeshGenerator",
    "ProcessGroupManager",
    "init_process_groups",
    "load_config",
    "merge_configs",
    "ParallelismConfig",
    "TrainingConfig",
    "Send",
    "Recv",
    "All_Gather",
    "All_Reduce",
    "ReduceScatter",
    "copy_to_group",
    "scatter_to_sequence",
    "pipeline_communicate",
    "bidirectional_pipeline_communicate",
    "get_rank",
    "get_world_size",
    "get_local_rank",
    "is_main_process",
    "setup_distributed",
    "cleanup_distributed",
    "barrier",
]
