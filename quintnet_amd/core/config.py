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

__all__ = ["load_config", "merge_configs", "validate_config", "ParallelismConfig", "TrainingConfig"]


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


_KNOWN_AXES = {"dp", "tp", "pp", "cp", "ep"}
_STRATEGIES = {"single_device", "ddp", "tensor_parallel", "pipeline_parallel",
               "dp_tp", "dp_pp", "3d"}


def validate_config(cfg: Dict[str, Any], world_size: Optional[int] = None) -> List[str]:
    """Early, readable validation of a loaded YAML config: returns a
    list of problem strings (empty = OK).  Catches the mistakes that
    otherwise surface as opaque rank crashes minutes into a run:
    mesh/world mismatch, unknown axis names, divisibility violations
    (heads, TP shards, expert placement), schedule typos."""
    errs: List[str] = []
    mesh = cfg.get("mesh_dim") or []
    names = cfg.get("mesh_name") or []
    if mesh and names and len(mesh) != len(names):
        errs.append(f"mesh_dim {mesh} and mesh_name {names} length mismatch")
    for n in names:
        if n not in _KNOWN_AXES:
            errs.append(f"unknown mesh axis {n!r} (known: {sorted(_KNOWN_AXES)})")
    if len(set(names)) != len(names):
        errs.append(f"duplicate axis in mesh_name {names}")
    for d in mesh:
        if not isinstance(d, int) or d < 1:
            errs.append(f"mesh_dim entries must be positive ints, got {mesh}")
            break
    if world_size is not None and mesh:
        prod = 1
        for d in mesh:
            prod *= d
        if prod != world_size:
            errs.append(f"mesh_dim {mesh} needs {prod} ranks, world is {world_size}")
    strat = cfg.get("strategy_name")
    if strat and strat not in _STRATEGIES:
        errs.append(f"unknown strategy {strat!r} (known: {sorted(_STRATEGIES)})")
    sched = cfg.get("schedule")
    if sched and sched not in {"afab", "1f1b", "interleaved_1f1b"}:
        errs.append(f"unknown pipeline schedule {sched!r}")
    lr_s = cfg.get("lr_schedule")
    if lr_s and lr_s not in {"constant", "linear", "cosine"}:
        errs.append(f"unknown lr_schedule {lr_s!r}")

    mc = cfg.get("model_config", {}) or {}
    axis = dict(zip(names, mesh))
    n_embd, n_head = mc.get("n_embd"), mc.get("n_head")
    if n_embd and n_head and n_embd % n_head:
        errs.append(f"n_embd {n_embd} not divisible by n_head {n_head}")
    tp = axis.get("tp", 1)
    if n_head and tp > 1 and n_head % tp:
        errs.append(f"n_head {n_head} not divisible by tp {tp}")
    if mc.get("n_inner") and tp > 1 and mc["n_inner"] % tp:
        errs.append(f"n_inner {mc['n_inner']} not divisible by tp {tp}")
    n_exp = mc.get("n_experts", 0)
    ep = axis.get("ep", 1)
    if n_exp and ep > 1 and n_exp % ep:
        errs.append(f"n_experts {n_exp} not divisible by ep {ep}")
    n_layer, pp = mc.get("n_layer"), axis.get("pp", 1)
    if n_layer and pp > 1 and n_layer < pp:
        errs.append(f"n_layer {n_layer} < pp {pp}: some stages would be empty")
    cp = axis.get("cp", 1)
    n_pos = mc.get("n_positions")
    if n_pos and cp > 1 and n_pos % (2 * cp):
        errs.append(f"n_positions {n_pos} not divisible by 2*cp={2*cp} "
                    "(zigzag CP needs 2 chunks per rank)")
    if mc.get("sequence_parallel") and cp > 1:
        errs.append("sequence_parallel and cp are exclusive")
    bs, acc = cfg.get("batch_size"), cfg.get("grad_acc_steps")
    if bs is not None and bs < 1:
        errs.append("batch_size must be >= 1")
    if acc is not None and acc < 1:
        errs.append("grad_acc_steps must be >= 1")
    return errs
