from .data_parallel import (
    BucketConfig,
    DataParallel,
    DistributedConfig,
    GradientBucket,
    create_distributed_ddp,
    create_local_ddp,
)
from .backends import DistributedBackend, TorchDistributedBackend, LocalBackend
from .tensor_parallel import (
    ColumnParallelLinear,
    RowParallelLinear,
    VocabParallelEmbedding,
    apply_tensor_parallel,
    ensure_divisibility,
)
from .context_parallel import (
    context_parallel_attention,
    zigzag_ring_attention,
    zigzag_to_context,
    zigzag_positions,
    zigzag_clm_targets,
    cp_causal_lm_loss,
    ring_attention,
    scatter_clm_targets,
    scatter_to_context,
)
from .expert_parallel import ExpertParallelMLP, all_to_all_var
from .zero3 import ZeRO3Block, apply_zero3
from .pipeline import (
    InterleavedPipelineWrapper,
    PipelineParallelWrapper,
    PipelineSchedule,
    AllFwdAllBwdSchedule,
    OneFOneBSchedule,
    PipelineTrainer,
    PipelineDataLoader,
    distribute_layers,
)

# Reference-compatible alias (QuintNet exports TensorParallel = apply_tensor_parallel)
TensorParallel = apply_tensor_parallel

__all__ = [
    "create_local_ddp",
    "create_distributed_ddp",
    "ExpertParallelMLP",
    "all_to_all_var",
    "cp_causal_lm_loss",
    "scatter_clm_targets",
    "ring_attention",
    "zigzag_ring_attention",
    "zigzag_to_context",
    "zigzag_positions",
    "zigzag_clm_targets",
    "ZeRO3Block",
    "apply_zero3",
    "context_parallel_attention",
    "scatter_to_context",
    "InterleavedPipelineWrapper",
    "DataParallel",
    "BucketConfig",
    "DistributedConfig",
    "GradientBucket",
    "DistributedBackend",
    "TorchDistributedBackend",
    "LocalBackend",
    "ColumnParallelLinear",
    "RowParallelLinear",
    "VocabParallelEmbedding",
    "apply_tensor_parallel",
    "TensorParallel",
    "ensure_divisibility",
    "PipelineParallelWrapper",
    "PipelineSchedule",
    "AllFwdAllBwdSchedule",
    "OneFOneBSchedule",
    "PipelineTrainer",
    "PipelineDataLoader",
    "distribute_layers",
]
