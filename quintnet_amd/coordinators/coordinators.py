"""The seven coordinators (reference coordinators/*.py).

Composition order is always TP (innermost) → PP → DP (outermost), each
engine receiving its named mesh subgroup.
"""

from __future__ import annotations

from typing import Optional

import torch.nn as nn

from ..parallel import (
    BucketConfig,
    DataParallel,
    DistributedConfig,
    PipelineParallelWrapper,
    apply_tensor_parallel,
)
from .base import BaseCoordinator

__all__ = [
    "DataParallelCoordinator",
    "TensorParallelCoordinator",
    "PipelineParallelCoordinator",
    "DPTCoordinator",
    "DPPCoordinator",
    "TPPCoordinator",
    "Hybrid3DCoordinator",
]


def _dp_wrap(model: nn.Module, pg, config=None) -> nn.Module:
    if pg.dp_size <= 1:
        return model
    cfg = DistributedConfig(
        rank=pg.dp_rank, world_size=pg.dp_size, process_group=pg.get_group("dp")
    )
    # zero_stage 2: reduce-scatter grad buckets (pair with optim.Zero2AdamW)
    bc = None
    if int((config or {}).get("zero_stage", 1)) == 2:
        bc = BucketConfig(grad_reduce_op="reduce_scatter")
    return DataParallel(model, config=cfg, bucket_config=bc)


def _tp_apply(model: nn.Module, pg, device) -> nn.Module:
    if pg.tp_size <= 1:
        return model.to(device)
    return apply_tensor_parallel(
        model,
        tp_size=pg.tp_size,
        tp_rank=pg.tp_rank,
        tp_group=pg.get_group("tp"),
        device=device,
        mode="column",
    )


def _pp_wrap(model: nn.Module, pg, device, stage_module=None, config=None) -> nn.Module:
    if pg.pp_size <= 1 and stage_module is None:
        return model.to(device)
    sched = str((config or {}).get("schedule", "1f1b")).lower()
    if sched in ("interleaved", "interleaved_1f1b", "vpp") and stage_module is None:
        from ..parallel import InterleavedPipelineWrapper

        return InterleavedPipelineWrapper(
            model,
            pp_rank=pg.pp_rank,
            pp_group=pg.get_group("pp") if "pp" in pg.mesh_name else None,
            pp_size=pg.pp_size,
            num_chunks=int((config or {}).get("num_chunks", 2)),
            device=device,
            tied_group=pg.get_tied_embedding_group()
            if pg.pp_rank in (0, pg.pp_size - 1)
            else None,
        )
    return PipelineParallelWrapper(
        model=model,
        pp_rank=pg.pp_rank,
        pp_group=pg.get_group("pp") if "pp" in pg.mesh_name else None,
        pp_size=pg.pp_size,
        device=device,
        stage_module=stage_module,
    )


class DataParallelCoordinator(BaseCoordinator):
    def parallelize(self) -> nn.Module:
        self.model.to(self.device)
        return _dp_wrap(self.model, self.pg, self.config)


class TensorParallelCoordinator(BaseCoordinator):
    def parallelize(self) -> nn.Module:
        return _tp_apply(self.model, self.pg, self.device)


class PipelineParallelCoordinator(BaseCoordinator):
    def parallelize(self) -> nn.Module:
        return _pp_wrap(self.model, self.pg, self.device, config=self.config)


class DPTCoordinator(BaseCoordinator):
    """TP then DP — with the dp subgroup (not the global group)."""

    def parallelize(self) -> nn.Module:
        m = _tp_apply(self.model, self.pg, self.device)
        return _dp_wrap(m, self.pg, self.config)


class DPPCoordinator(BaseCoordinator):
    def parallelize(self) -> nn.Module:
        m = _pp_wrap(self.model, self.pg, self.device, config=self.config)
        return _dp_wrap(m, self.pg, self.config)


class TPPCoordinator(BaseCoordinator):
    def parallelize(self) -> nn.Module:
        m = _tp_apply(self.model, self.pg, self.device)
        return _pp_wrap(m, self.pg, self.device, config=self.config)


class Hybrid3DCoordinator(BaseCoordinator):
    """TP → PP → DP; optional staged checkpoint path for GPT-2.

    Reference parity: coordinators/hybrid_3d_coordinator.py:49-236.
    """

    def __init__(self, model, pg_manager, config=None, checkpoint_path=None, is_staged=False, **kw):
        super().__init__(model, pg_manager, config, **kw)
        self.checkpoint_path = checkpoint_path
        self.is_staged = is_staged

    def parallelize(self) -> nn.Module:
        if self.is_staged:
            return self._parallelize_staged()
        return self._parallelize_non_staged()

    def _parallelize_non_staged(self) -> nn.Module:
        m = _tp_apply(self.model, self.pg, self.device)
        m = _pp_wrap(m, self.pg, self.device, config=self.config)
        return _dp_wrap(m, self.pg, self.config)

    def _parallelize_staged(self) -> nn.Module:
        from ..checkpoint.distributed_loading import load_gpt2_distributed
        from ..models.gpt2 import GPT2Config, GPT2Stage

        mc = self.config.get("model_config", {})
        gcfg = GPT2Config(
            vocab_size=mc.get("vocab_size", 50257),
            n_positions=mc.get("n_positions", 1024),
            n_embd=mc.get("n_embd", 768),
            n_layer=mc.get("n_layer", 12),
            n_head=mc.get("n_head", 12),
            n_inner=mc.get("n_inner", None),
            dropout=mc.get("dropout", 0.1),
        )
        state = load_gpt2_distributed(
            self.checkpoint_path,
            gcfg,
            pp_rank=self.pg.pp_rank,
            pp_size=self.pg.pp_size,
            tp_rank=self.pg.tp_rank,
            tp_size=self.pg.tp_size,
        )
        stage = GPT2Stage.from_sharded_state_dict(
            gcfg,
            state,
            pp_rank=self.pg.pp_rank,
            pp_size=self.pg.pp_size,
            tp_group=self.pg.get_group("tp") if "tp" in self.pg.mesh_name else None,
            tied_group=self.pg.get_tied_embedding_group(),
            device=self.device,
        )
        m = _pp_wrap(None, self.pg, self.device, stage_module=stage)
        return _dp_wrap(m, self.pg, self.config)
