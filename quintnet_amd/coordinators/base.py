"""BaseCoordinator (reference coordinators/main_coordinator.py:36-70)."""

from __future__ import annotations

import abc
from typing import Any, Dict, Optional

import torch
import torch.nn as nn

__all__ = ["BaseCoordinator"]


class BaseCoordinator(abc.ABC):
    def __init__(
        self,
        model: Optional[nn.Module],
        pg_manager,
        config: Optional[Dict[str, Any]] = None,
        **kwargs,
    ):
        self.model = model
        self.pg = pg_manager
        self.config = config or {}
        self.kwargs = kwargs

    @property
    def device(self) -> torch.device:
        return self.pg.device

    @abc.abstractmethod
    def parallelize(self) -> nn.Module:
        ...
