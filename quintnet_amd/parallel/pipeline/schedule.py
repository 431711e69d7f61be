"""Pipeline schedules: 1F1B and AFAB over RCCL P2P.

Parity with reference parallelism/pipeline_parallel/schedule.py:74-516
(AllFwdAllBwdSchedule, OneFOneBSchedule; classification + clm task
types), redesigned around statically-known tensor shapes and
event-ordered P2P (no device-wide synchronize — see core/comm.py).

DP interplay: gradients are accumulated locally across micro-batches;
the DDP bucket all-reduce fires only on the final backward micro-batch
(``require_backward_grad_sync``), overlapping the tail of backward —
this is the *intended* semantics the reference never reached (SURVEY.md
§8.1-8.2).
"""

from __future__ import annotations

import abc
from typing import Any, Dict, List, Optional, Tuple

import torch

from ...core.comm import (
    bidirectional_pipeline_communicate,
    pipeline_communicate,
)
from ...ops import causal_lm_loss, clip_grad_norm_local, shift_labels

__all__ = ["PipelineSchedule", "AllFwdAllBwdSchedule", "OneFOneBSchedule", "get_schedule"]


class PipelineSchedule(abc.ABC):
    def __init__(
        self,
        trainer,
        task_type: str = "classification",
    ):
        self.trainer = trainer
        self.task_type = task_type

    # -- conveniences over the owning PipelineTrainer -------------------
    @property
    def model(self):
        return self.trainer.model

    @property
    def pp_rank(self) -> int:
        return self.trainer.pp_rank

    @property
    def pp_size(self) -> int:
        return self.trainer.pp_size

    @property
    def pp_group(self):
        return self.trainer.pp_group

    @property
    def group_ranks(self) -> List[int]:
        return self.trainer.pp_group_ranks

    @property
    def is_first(self) -> bool:
        return self.pp_rank == 0

    @property
    def is_last(self) -> bool:
        return self.pp_rank == self.pp_size - 1

    # ------------------------------------------------------------------
    def _set_grad_sync(self, enabled: bool) -> None:
        m = self.model
        if hasattr(m, "require_backward_grad_sync"):
            m.require_backward_grad_sync = enabled

    def _finalize_grads(self) -> None:
        m = self.model
        if hasattr(m, "finalize_gradients"):
            m.finalize_gradients()
        # find the tied-weight owner through DataParallel(.module) /
        # PipelineParallelWrapper(.local_module) nesting
        obj = m
        for _ in range(4):
            if hasattr(obj, "sync_tied_weights_grad"):
                obj.sync_tied_weights_grad()
                return
            if hasattr(obj, "module"):
                obj = obj.module
            elif hasattr(obj, "local_module"):
                obj = obj.local_module
            else:
                return

    def _stage_input(self, batch: Dict[str, Any], device) -> torch.Tensor:
        if self.task_type == "clm":
            return batch["input_ids"].to(device, non_blocking=True)
        return batch["images"].to(device, non_blocking=True)

    def _loss_and_metrics(
        self, output: torch.Tensor, batch: Dict[str, Any], device, metrics: Dict[str, float]
    ) -> torch.Tensor:
        labels = batch["labels"].to(device, non_blocking=True)
        if self.task_type == "clm":
            targets = shift_labels(labels, -100)
            loss = causal_lm_loss(output, labels, ignore_index=-100)
            with torch.no_grad():
                n_tok = int((targets != -100).sum())
                metrics["loss"] = metrics.get("loss", 0.0) + float(loss.detach())
                metrics["n_tokens"] = metrics.get("n_tokens", 0) + n_tok
        else:
            loss = self.trainer.criterion(output, labels)
            with torch.no_grad():
                pred = output.argmax(dim=-1)
                metrics["loss"] = metrics.get("loss", 0.0) + float(loss.detach())
                metrics["correct"] = metrics.get("correct", 0) + int((pred == labels).sum())
                metrics["total"] = metrics.get("total", 0) + labels.numel()
        return loss

    def _forward_step(
        self,
        data_loader,
        input_tensor: Optional[torch.Tensor],
        device,
        metrics: Dict[str, float],
        num_micro: int,
    ) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
        """Run one micro-batch forward.

        Returns (output_for_send_or_loss, stage_input_tensor).  Every
        stage consumes the loader to stay in lockstep (labels come from
        the same batch object on the last stage — SURVEY.md §8.6).
        """
        batch = next(data_loader)
        if self.is_first:
            x = self._stage_input(batch, device)
        else:
            x = input_tensor
        out = self.model.forward(x)
        if self.is_last:
            loss = self._loss_and_metrics(out, batch, device, metrics)
            return loss / num_micro, input_tensor
        return out, input_tensor

    def _backward_step(
        self,
        input_tensor: Optional[torch.Tensor],
        output_tensor: torch.Tensor,
        output_grad: Optional[torch.Tensor],
    ) -> Optional[torch.Tensor]:
        return self.model.backward(input_tensor, output_tensor, output_grad)

    def _optimizer_step(self) -> None:
        t = self.trainer
        self._finalize_grads()
        if t.max_grad_norm is not None and t.max_grad_norm > 0:
            if hasattr(t.optimizer, "clip_grad_norm_"):
                t.optimizer.clip_grad_norm_(t.max_grad_norm)  # flat-buffer clip
            else:
                clip_grad_norm_local(
                    [p for p in self.model.parameters() if p.requires_grad],
                    t.max_grad_norm,
                )
        t.optimizer.step()
        if hasattr(self.model, "zero_grad"):
            self.model.zero_grad()
        else:
            t.optimizer.zero_grad()

    @abc.abstractmethod
    def train_step(self, data_loader, tensor_shapes, device, dtype) -> Dict[str, float]:
        ...

    def _final_metrics(self, metrics: Dict[str, float], num_micro: int) -> Dict[str, float]:
        out: Dict[str, float] = {}
        if self.is_last:
            out["loss"] = metrics.get("loss", 0.0) / max(num_micro, 1)
            if self.task_type == "clm":
                out["n_tokens"] = metrics.get("n_tokens", 0)
                out["ppl"] = float(torch.exp(torch.tensor(min(out["loss"], 20.0))))
            else:
                tot = max(metrics.get("total", 0), 1)
                out["accuracy"] = 100.0 * metrics.get("correct", 0) / tot
        return out


class AllFwdAllBwdSchedule(PipelineSchedule):
    """AFAB: all forwards, then all backwards (reference schedule.py:74-246)."""

    def train_step(self, data_loader, tensor_shapes, device, dtype) -> Dict[str, float]:
        num_micro = data_loader.grad_acc_steps
        metrics: Dict[str, float] = {}
        inputs: List[Optional[torch.Tensor]] = []
        outputs: List[torch.Tensor] = []
        self._set_grad_sync(False)

        for _ in range(num_micro):
            inp = pipeline_communicate(
                "recv_forward", self.pp_rank, self.pp_size, self.group_ranks,
                shapes=tensor_shapes, dtype=dtype, device=device, group=self.pp_group,
            )
            out, inp = self._forward_step(data_loader, inp, device, metrics, num_micro)
            if not self.is_last:
                pipeline_communicate(
                    "send_forward", self.pp_rank, self.pp_size, self.group_ranks,
                    tensor=out, group=self.pp_group,
                )
            inputs.append(inp)
            outputs.append(out)

        for i in range(num_micro):
            if i == num_micro - 1:
                self._set_grad_sync(True)
            grad = pipeline_communicate(
                "recv_backward", self.pp_rank, self.pp_size, self.group_ranks,
                shapes=tensor_shapes, dtype=dtype, device=device, group=self.pp_group,
            )
            in_grad = self._backward_step(inputs[i], outputs[i], grad)
            if not self.is_first:
                pipeline_communicate(
                    "send_backward", self.pp_rank, self.pp_size, self.group_ranks,
                    tensor=in_grad, group=self.pp_group,
                )

        self._optimizer_step()
        return self._final_metrics(metrics, num_micro)


class OneFOneBSchedule(PipelineSchedule):
    """1F1B: warmup fwd, steady 1F1B with bidirectional P2P, cooldown bwd.

    Reference schedule.py:257-516.
    """

    def train_step(self, data_loader, tensor_shapes, device, dtype) -> Dict[str, float]:
        num_micro = data_loader.grad_acc_steps
        warmup = min(self.pp_size - self.pp_rank - 1, num_micro)
        remaining = num_micro - warmup
        metrics: Dict[str, float] = {}
        inputs: List[Optional[torch.Tensor]] = []
        outputs: List[torch.Tensor] = []
        backwards_done = 0
        self._set_grad_sync(False)

        def maybe_enable_sync():
            nonlocal backwards_done
            backwards_done += 1
            if backwards_done == num_micro:
                pass  # sync was enabled just before this backward

        # -- warmup forwards --------------------------------------------
        for _ in range(warmup):
            inp = pipeline_communicate(
                "recv_forward", self.pp_rank, self.pp_size, self.group_ranks,
                shapes=tensor_shapes, dtype=dtype, device=device, group=self.pp_group,
            )
            out, inp = self._forward_step(data_loader, inp, device, metrics, num_micro)
            if not self.is_last:
                pipeline_communicate(
                    "send_forward", self.pp_rank, self.pp_size, self.group_ranks,
                    tensor=out, group=self.pp_group,
                )
            inputs.append(inp)
            outputs.append(out)

        # -- steady state -----------------------------------------------
        if remaining > 0:
            inp = pipeline_communicate(
                "recv_forward", self.pp_rank, self.pp_size, self.group_ranks,
                shapes=tensor_shapes, dtype=dtype, device=device, group=self.pp_group,
            )
        for i in range(remaining):
            out, inp_used = self._forward_step(data_loader, inp, device, metrics, num_micro)
            if self.is_last:
                out_grad = None
            else:
                out_grad = bidirectional_pipeline_communicate(
                    "send_fwd_recv_bwd", self.pp_rank, self.pp_size, self.group_ranks,
                    send_tensor=out, recv_shapes=tensor_shapes, dtype=dtype,
                    device=device, group=self.pp_group,
                )
            inputs.append(inp_used)
            outputs.append(out)
            b_in, b_out = inputs.pop(0), outputs.pop(0)
            if backwards_done + 1 == num_micro:
                self._set_grad_sync(True)
            in_grad = self._backward_step(b_in, b_out, out_grad)
            maybe_enable_sync()
            is_last_iter = i == remaining - 1
            if self.is_first:
                inp = (
                    None
                    if is_last_iter
                    else pipeline_communicate(
                        "recv_forward", self.pp_rank, self.pp_size, self.group_ranks,
                        shapes=tensor_shapes, dtype=dtype, device=device, group=self.pp_group,
                    )
                )
            elif is_last_iter:
                pipeline_communicate(
                    "send_backward", self.pp_rank, self.pp_size, self.group_ranks,
                    tensor=in_grad, group=self.pp_group,
                )
                inp = None
            else:
                inp = bidirectional_pipeline_communicate(
                    "send_bwd_recv_fwd", self.pp_rank, self.pp_size, self.group_ranks,
                    send_tensor=in_grad, recv_shapes=tensor_shapes, dtype=dtype,
                    device=device, group=self.pp_group,
                )

        # -- cooldown backwards -----------------------------------------
        for _ in range(warmup):
            grad = pipeline_communicate(
                "recv_backward", self.pp_rank, self.pp_size, self.group_ranks,
                shapes=tensor_shapes, dtype=dtype, device=device, group=self.pp_group,
            )
            b_in, b_out = inputs.pop(0), outputs.pop(0)
            if backwards_done + 1 == num_micro:
                self._set_grad_sync(True)
            in_grad = self._backward_step(b_in, b_out, grad)
            maybe_enable_sync()
            if not self.is_first:
                pipeline_communicate(
                    "send_backward", self.pp_rank, self.pp_size, self.group_ranks,
                    tensor=in_grad, group=self.pp_group,
                )

        self._optimizer_step()
        return self._final_metrics(metrics, num_micro)


def get_schedule(name: str, trainer, task_type: str) -> PipelineSchedule:
    name = (name or "1f1b").lower()
    if name in ("1f1b", "one_f_one_b", "onefoneb"):
        return OneFOneBSchedule(trainer, task_type)
    if name in ("afab", "all_forward_all_backward", "gpipe"):
        return AllFwdAllBwdSchedule(trainer, task_type)
    raise ValueError(f"unknown schedule {name!r}")
