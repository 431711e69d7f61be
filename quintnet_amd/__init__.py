"""quintnet_amd — MI355X-native 3D-parallel training framework.

A from-scratch rebuild of the QuintNet capability set (device mesh +
process groups for [dp, tp, pp]; bucketed DP; Megatron TP; 1F1B/AFAB PP;
7 composable strategies; ViT + GPT-2 model families; trainers; sharded
checkpoints + merge; staged loading) designed for AMD Instinct MI355X:
PyTorch-ROCm orchestration, hand-written HIP/CDNA4 (gfx950) kernels for
the hot ops, and RCCL collectives over xGMI.  Beyond reference parity:
interleaved 1F1B, sequence/context/expert parallelism, ZeRO-3,
KV-cached generation, fp8 forward path, per-phase tracing, watchdog.

Public API parity with the reference package root (__init__.py:17-37).
"""

__version__ = "0.2.0"

from .core import (
    init_process_groups,
    ProcessGroupManager,
    MeshGenerator,
    load_config,
    Send,
    Recv,
    All_Gather,
    All_Reduce,
    ReduceScatter,
    pipeline_communicate,
    bidirectional_pipeline_communicate,
)
from .strategy import get_strategy
from .trainer import Trainer
from .gpt2_trainer import GPT2Trainer
from .parallel import (
    DataParallel,
    TensorParallel,
    apply_tensor_parallel,
    ColumnParallelLinear,
    RowParallelLinear,
    VocabParallelEmbedding,
    PipelineParallelWrapper,
    InterleavedPipelineWrapper,
    PipelineTrainer,
    PipelineDataLoader,
    context_parallel_attention,
    ring_attention,
    ExpertParallelMLP,
    ZeRO3Block,
    apply_zero3,
)
from .optim import ZeroRedundancyAdamW, DistributedAdamW

__all__ = [
    "init_process_groups",
    "ProcessGroupManager",
    "MeshGenerator",
    "load_config",
    "Send",
    "Recv",
    "All_Gather",
    "All_Reduce",
    "ReduceScatter",
    "pipeline_communicate",
    "bidirectional_pipeline_communicate",
    "get_strategy",
    "Trainer",
    "GPT2Trainer",
    "DataParallel",
    "TensorParallel",
    "apply_tensor_parallel",
    "ColumnParallelLinear",
    "RowParallelLinear",
    "VocabParallelEmbedding",
    "PipelineParallelWrapper",
    "InterleavedPipelineWrapper",
    "PipelineTrainer",
    "PipelineDataLoader",
    "context_parallel_attention",
    "ring_attention",
    "ExpertParallelMLP",
    "ZeRO3Block",
    "apply_zero3",
    "ZeroRedundancyAdamW",
    "DistributedAdamW",
]
