"""YAML config loading (reference parity: core/config.py:96-130).

Keeps the reference's key contract: batch_size,
num_epochs, learning_rate, grad_acc_steps, max_grad_norm, num_workers,
dataset_path, model dims, and parallelism keys (device_type, mesh_dim,
mesh_name, strategy_name, schedule).
"""

from __future__ import annotations

import dataclasses
from typing import Any, Dict, List, Optional

import yaml

__all__ = ["load_config", "merge_configs", "ParallelismConfig", "TrainingConfig"]


@dataclasses.dataclass
class ParallelismConfig:
    device_type: str = "cuda"
    mesh_dim: List[int] = dataclasses.field(default_factory=lambda: [2, 2, 2])
    mesh_name: List[str] = dataclasses.field(default_factory=lambda: ["dp", "tp", "pp"])
    strategy_name: str = "3d"
    schedule: str = "1f1b"
    timeout_s: float = 600.0


@dataclasses.dataclass
class TrainingConfig:
    batch_size: int = 32
    num_epochs: int = 1
    learning_rate: float = 1e-3
    grad_acc_steps: int = 1
    max_grad_norm: float = 1.0
    num_workers: int = 0
    dataset_path: Optional[str] = None


def load_config(path: str) -> Dict[str, Any]:
    with open(path, "r") as f:
        cfg = yaml.safe_load(f)
    if cfg is None:
        cfg = {}
    if not isinstance(cfg, dict):
        raise ValueError(f"config {path} did not parse to a mapping")
    return cfg


def merge_configs(base: Dict[str, Any], override: Dict[str, Any]) -> Dict[str, Any]:
    """Deep-merge ``override`` into ``base`` (reference left this a stub)."""
    out = dict(base)
    for k, v in override.items():
        if k in out and isinstance(out[k], dict) and isinstance(v, dict):
            out[k] = merge_configs(out[k], v)
        else:
            out[k] = v
    return out
