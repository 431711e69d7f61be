"""PipelineTrainer: per-step driver owning optimizer/criterion/schedule.

Parity with reference parallelism/pipeline_parallel/trainer.py:105-281.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch

from ...core.comm import pipeline_communicate
from .schedule import get_schedule

__all__ = ["PipelineTrainer"]


class PipelineTrainer:
    def __init__(
        self,
        model,
        optimizer,
        criterion,
        pp_rank: int,
        pp_size: int,
        pp_group,
        pp_group_ranks: List[int],
        schedule: str = "1f1b",
        task_type: str = "classification",
        max_grad_norm: Optional[float] = 1.0,
        pp_fwd_group=None,
        pp_bwd_group=None,
        cp_group=None,
        tp_group=None,
        defer_wgrads: bool = False,
    ):
        self.model = model
        self.pp_fwd_group = pp_fwd_group
        self.pp_bwd_group = pp_bwd_group
        self.cp_group = cp_group
        self.tp_group = tp_group  # for the global grad-norm clip
        self.optimizer = optimizer
        self.criterion = criterion
        self.pp_rank = pp_rank
        self.pp_size = pp_size
        self.pp_group = pp_group
        self.pp_group_ranks = pp_group_ranks
        self.max_grad_norm = max_grad_norm
        self.task_type = task_type
        self.lr_scheduler = None  # set by the owning Trainer at fit()
        self.defer_wgrads = bool(defer_wgrads)
        self.schedule = get_schedule(schedule, self, task_type)

    def _lr_step(self) -> None:
        if self.lr_scheduler is not None:
            self.lr_scheduler.step()

    @property
    def is_last_stage(self) -> bool:
        return self.pp_rank == self.pp_size - 1

    def train_step(self, data_loader, tensor_shapes, device, dtype=torch.float32) -> Dict[str, float]:
        return self.schedule.train_step(data_loader, tensor_shapes, device, dtype)

    @torch.no_grad()
    def evaluate(self, data_loader, tensor_shapes, device, dtype=torch.float32, max_batches=None) -> Dict[str, float]:
        """Forward-only pipeline loop; last stage computes metrics."""
        metrics: Dict[str, float] = {}
        n = 0
        sched = self.schedule
        from .dataloader import PipelineDataLoader

        from .schedule import InterleavedOneFOneBSchedule

        if isinstance(sched, InterleavedOneFOneBSchedule):
            return self._evaluate_interleaved(
                data_loader, tensor_shapes, device, dtype, max_batches
            )

        for batch in data_loader:
            if max_batches is not None and n >= max_batches:
                break
            batch = PipelineDataLoader._normalize(batch)
            n += 1
            if sched.is_first:
                x = sched._stage_input(batch, device)
            else:
                x = pipeline_communicate(
                    "recv_forward", self.pp_rank, self.pp_size, self.pp_group_ranks,
                    shapes=tensor_shapes, dtype=dtype, device=device, group=self.pp_group,
                )
            out = self.model(x)
            if sched.is_last:
                sched._loss_and_metrics(out, batch, device, metrics)
            else:
                pipeline_communicate(
                    "send_forward", self.pp_rank, self.pp_size, self.pp_group_ranks,
                    tensor=out, group=self.pp_group,
                )
        return sched._final_metrics(metrics, n)

    @torch.no_grad()
    def _evaluate_interleaved(self, data_loader, tensor_shapes, device, dtype,
                              max_batches=None) -> Dict[str, float]:
        """Forward-only ring walk over the v*p virtual stages per batch."""
        from ...core.comm import ring_recv, ring_send
        from .dataloader import PipelineDataLoader

        sched = self.schedule
        owner = sched._chunks_owner()
        v, p, r = int(owner.num_chunks), self.pp_size, self.pp_rank
        fwd_group = self.pp_fwd_group or self.pp_group
        metrics: Dict[str, float] = {}
        n = 0
        for batch in data_loader:
            if max_batches is not None and n >= max_batches:
                break
            batch = PipelineDataLoader._normalize(batch)
            n += 1
            for g in range(v * p):
                if g % p != r:
                    continue
                if g == 0:
                    inp = sched._stage_input(batch, device)
                else:
                    inp = ring_recv(r, p, self.pp_group_ranks, -1, tensor_shapes,
                                    dtype, device, group=fwd_group)
                out = owner.forward(inp, chunk_id=g // p)
                if g == v * p - 1:
                    sched._loss_and_metrics(out, batch, device, metrics)
                else:
                    for rq in ring_send(out, r, p, self.pp_group_ranks, +1,
                                        group=fwd_group):
                        rq.wait()
        return sched._final_metrics(metrics, n)
