from .logger import setup_rank_logger, print_rank_0
from .memory import memory_stats, print_memory_stats
from .profiling import StepTimer, PhaseTimer
from .watchdog import Watchdog
from .graphs import CapturedStep
from .metrics import (
    rouge_n,
    rouge_l,
    bleu,
    compute_generation_metrics,
    generate_greedy,
    count_parameters,
)
from .data import (
    CustomDataset,
    mnist_transform,
    SyntheticMNIST,
    SyntheticCLM,
    SummarizationDataset,
    SummarizationCollator,
    SummarizationDataLoader,
)

__all__ = [
    "setup_rank_logger",
    "print_rank_0",
    "memory_stats",
    "print_memory_stats",
    "StepTimer",
    "PhaseTimer",
    "Watchdog",
    "CapturedStep",
    "rouge_n",
    "rouge_l",
    "bleu",
    "compute_generation_metrics",
    "generate_greedy",
    "count_parameters",
    "CustomDataset",
    "mnist_transform",
    "SyntheticMNIST",
    "SyntheticCLM",
    "SummarizationDataset",
    "SummarizationCollator",
    "SummarizationDataLoader",
]
