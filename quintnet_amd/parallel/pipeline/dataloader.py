"""Infinite wrapping micro-batch iterator for pipeline schedules.

Parity with reference parallelism/pipeline_parallel/dataloader.py:17-56.
"""

from __future__ import annotations

from typing import Any, Dict, Iterator

__all__ = ["PipelineDataLoader"]


class PipelineDataLoader:
    def __init__(self, dataloader, grad_acc_steps: int = 1, task_type: str = None):
        self.dataloader = dataloader
        self.grad_acc_steps = grad_acc_steps
        self._iter: Iterator = iter(dataloader)
        self.task_type = task_type or self._detect_task_type()

    def _detect_task_type(self) -> str:
        try:
            batch = next(iter(self.dataloader))
        except StopIteration:
            return "classification"
        if isinstance(batch, dict) and ("input_ids" in batch or "labels" in batch and "attention_mask" in batch):
            return "clm"
        return "classification"

    def __len__(self) -> int:
        return len(self.dataloader)

    def __iter__(self):
        return self

    def __next__(self) -> Dict[str, Any]:
        try:
            batch = next(self._iter)
        except StopIteration:
            self._iter = iter(self.dataloader)
            batch = next(self._iter)
        return self._normalize(batch)

    @staticmethod
    def _normalize(batch) -> Dict[str, Any]:
        if isinstance(batch, dict):
            out = dict(batch)
            if "image" in out:
                out["images"] = out.pop("image")
            if "label" in out:
                out["labels"] = out.pop("label")
            return out
        if isinstance(batch, (list, tuple)) and len(batch) == 2:
            return {"images": batch[0], "labels": batch[1]}
        raise ValueError(f"unsupported batch type {type(batch)}")
