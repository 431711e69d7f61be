"""Coordinators: ordered composition of parallelism engines (TP → PP → DP).

Parity with reference coordinators/ (main_coordinator.py:36-70 and the
seven per-strategy coordinators).  Every coordinator passes the NAMED
mesh subgroup to each engine — including DP+TP, where the reference
passed the default/global group and would have all-reduced gradients
across TP ranks too.
"""

from .base import BaseCoordinator
from .coordinators import (
    DataParallelCoordinator,
    TensorParallelCoordinator,
    PipelineParallelCoordinator,
    DPTCoordinator,
    DPPCoordinator,
    TPPCoordinator,
    Hybrid3DCoordinator,
)

__all__ = [
    "BaseCoordinator",
    "DataParallelCoordinator",
    "TensorParallelCoordinator",
    "PipelineParallelCoordinator",
    "DPTCoordinator",
    "DPPCoordinator",
    "TPPCoordinator",
    "Hybrid3DCoordinator",
]
