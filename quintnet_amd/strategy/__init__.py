"""Strategy facade: one-line model parallelization API.

Parity with reference strategy/ (get_strategy registry at
strategy/__init__.py:52-105; BaseStrategy.apply at base_strategy.py:71-84).
"""

from __future__ import annotations

import abc
import os
from typing import Any, Dict, Optional

import torch
import torch.nn as nn

from ..coordinators import (
    DataParallelCoordinator,
    DPPCoordinator,
    DPTCoordinator,
    Hybrid3DCoordinator,
    PipelineParallelCoordinator,
    TensorParallelCoordinator,
    TPPCoordinator,
)

__all__ = [
    "BaseStrategy",
    "DataParallelStrategy",
    "TensorParallelStrategy",
    "PipelineParallelStrategy",
    "DPTStrategy",
    "DPPStrategy",
    "TPPStrategy",
    "Hybrid3DStrategy",
    "get_strategy",
]


class BaseStrategy(abc.ABC):
    coordinator_cls = None

    def __init__(
        self,
        pg_manager,
        config: Optional[Dict[str, Any]] = None,
        checkpoint_path: Optional[str] = None,
        is_staged: bool = False,
    ):
        self.pg = pg_manager
        self.config = config or {}
        self.checkpoint_path = checkpoint_path
        self.is_staged = is_staged

    def _get_device(self) -> torch.device:
        if self.pg is not None:
            return self.pg.device
        if torch.cuda.is_available():
            return torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0)))
        return torch.device("cpu")

    def apply(self, model: Optional[nn.Module]) -> nn.Module:
        coord = self.coordinator_cls(model, self.pg, self.config)
        return coord.parallelize()


class DataParallelStrategy(BaseStrategy):
    coordinator_cls = DataParallelCoordinator


class TensorParallelStrategy(BaseStrategy):
    coordinator_cls = TensorParallelCoordinator


class PipelineParallelStrategy(BaseStrategy):
    coordinator_cls = PipelineParallelCoordinator


class DPTStrategy(BaseStrategy):
    coordinator_cls = DPTCoordinator


class DPPStrategy(BaseStrategy):
    coordinator_cls = DPPCoordinator


class TPPStrategy(BaseStrategy):
    coordinator_cls = TPPCoordinator


class Hybrid3DStrategy(BaseStrategy):
    def apply(self, model: Optional[nn.Module]) -> nn.Module:
        coord = Hybrid3DCoordinator(
            model,
            self.pg,
            self.config,
            checkpoint_path=self.checkpoint_path,
            is_staged=self.is_staged,
        )
        return coord.parallelize()


_REGISTRY = {
    "dp": DataParallelStrategy,
    "tp": TensorParallelStrategy,
    "pp": PipelineParallelStrategy,
    "dp_tp": DPTStrategy,
    "dp_pp": DPPStrategy,
    "tp_pp": TPPStrategy,
    "3d": Hybrid3DStrategy,
}


def get_strategy(
    name: str,
    pg_manager,
    config: Optional[Dict[str, Any]] = None,
    checkpoint_path: Optional[str] = None,
    is_staged: bool = False,
) -> BaseStrategy:
    name = name.lower()
    if name not in _REGISTRY:
        raise ValueError(f"unknown strategy {name!r}; choose from {sorted(_REGISTRY)}")
    return _REGISTRY[name](
        pg_manager, config, checkpoint_path=checkpoint_path, is_staged=is_staged
    )
