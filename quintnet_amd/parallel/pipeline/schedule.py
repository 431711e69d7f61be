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
from ...ops import causal_lm_loss, shift_labels

__all__ = [
    "PipelineSchedule",
    "AllFwdAllBwdSchedule",
    "OneFOneBSchedule",
    "InterleavedOneFOneBSchedule",
    "get_schedule",
]


class PipelineSchedule(abc.ABC):
    def __init__(
        self,
        trainer,
        task_type: str = "classification",
    ):
        self.trainer = trainer
        self.task_type = task_type
        # zero-bubble dW deferral (opt-in via trainer config; the
        # trainer object carries the flag from GPT2Trainer/Trainer)
        self._defer_wgrads = bool(getattr(trainer, "defer_wgrads", False))

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

    @property
    def cp_group(self):
        return getattr(self.trainer, "cp_group", None)

    def _stage_input(self, batch: Dict[str, Any], device) -> torch.Tensor:
        if self.task_type == "clm":
            ids = batch["input_ids"].to(device, non_blocking=True)
            if self.cp_group is not None:
                from ..context_parallel import scatter_to_context

                ids = scatter_to_context(ids, self.cp_group, dim=1)
            return ids
        return batch["images"].to(device, non_blocking=True)

    def _loss_and_metrics(
        self, output: torch.Tensor, batch: Dict[str, Any], device, metrics: Dict[str, float]
    ) -> torch.Tensor:
        labels = batch["labels"].to(device, non_blocking=True)
        if self.task_type == "clm" and self.cp_group is not None:
            from ..context_parallel import cp_causal_lm_loss, scatter_clm_targets

            tgt = scatter_clm_targets(labels, self.cp_group)
            loss, true_loss = cp_causal_lm_loss(output, tgt, self.cp_group)
            with torch.no_grad():
                metrics["loss"] = metrics.get("loss", 0.0) + float(true_loss)
                metrics["n_tokens"] = metrics.get("n_tokens", 0) + int(
                    (tgt != -100).sum()
                )
        elif self.task_type == "clm":
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

    @property
    def _pt(self):
        return getattr(self.trainer, "phase_timer", None)

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
        pt = self._pt
        if pt:
            pt.start("forward")
        try:
            return self._forward_step_inner(
                data_loader, input_tensor, device, metrics, num_micro
            )
        finally:
            if pt:
                pt.stop("forward")

    def _forward_step_inner(
        self,
        data_loader,
        input_tensor: Optional[torch.Tensor],
        device,
        metrics: Dict[str, float],
        num_micro: int,
    ) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
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
        pt = self._pt
        if pt:
            pt.start("backward")
        try:
            if self._defer_wgrads:
                # zero-bubble mode: the backward computes only dX (the
                # critical path feeding the upstream stage); the weight
                # GEMMs queue and run in flush_deferred_wgrads() at the
                # optimizer step — on GPU their kernels overlap whatever
                # the stage would otherwise idle on (ops/linear.py)
                from ...ops import defer_wgrads

                with defer_wgrads.scope():
                    return self.model.backward(
                        input_tensor, output_tensor, output_grad
                    )
            return self.model.backward(input_tensor, output_tensor, output_grad)
        finally:
            if pt:
                pt.stop("backward")

    def _optimizer_step(self) -> None:
        t = self.trainer
        pt = self._pt
        if pt:
            pt.start("optimizer")
        if self._defer_wgrads:
            from ...ops import flush_deferred_wgrads

            flush_deferred_wgrads()  # BEFORE reduction/tied-sync/clip
            # hooks stayed off for the whole window (see the gated
            # _set_grad_sync sites); enable now so finalize launches
            # every bucket itself with the flushed, complete gradients
            self._set_grad_sync(True)
        self._finalize_grads()
        if t.max_grad_norm is not None and t.max_grad_norm > 0:
            if hasattr(t.optimizer, "clip_grad_norm_"):
                t.optimizer.clip_grad_norm_(t.max_grad_norm)  # flat-buffer clip
            else:
                from ...ops import clip_grad_norm_global

                clip_grad_norm_global(
                    [p for p in self.model.parameters() if p.requires_grad],
                    t.max_grad_norm,
                    tp_group=getattr(t, "tp_group", None),
                    pp_group=self.pp_group,
                )
        if hasattr(t, "_lr_step"):
            t._lr_step()
        t.optimizer.step()
        if hasattr(self.model, "zero_grad"):
            self.model.zero_grad()
        else:
            t.optimizer.zero_grad()
        if pt:
            pt.stop("optimizer")

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
            if i == num_micro - 1 and not self._defer_wgrads:
                self._set_grad_sync(True)
            if self._defer_wgrads and i:
                from ...ops import flush_deferred_wgrads

                flush_deferred_wgrads()  # ZB: dW kernels under the recv wait
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
                if self._defer_wgrads:
                    from ...ops import flush_deferred_wgrads

                    # ZB: launch the queued dW GEMMs before blocking on
                    # the bwd-grad recv — their kernels fill the wait
                    flush_deferred_wgrads()
                out_grad = bidirectional_pipeline_communicate(
                    "send_fwd_recv_bwd", self.pp_rank, self.pp_size, self.group_ranks,
                    send_tensor=out, recv_shapes=tensor_shapes, dtype=dtype,
                    device=device, group=self.pp_group,
                )
            inputs.append(inp_used)
            outputs.append(out)
            b_in, b_out = inputs.pop(0), outputs.pop(0)
            if backwards_done + 1 == num_micro and not self._defer_wgrads:
                # (ZB: hooks stay off — post-accumulate hooks fire even
                # for deferred params, which would launch the bucket
                # reduction BEFORE the final flush; finalize launches)
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
            if self._defer_wgrads:
                from ...ops import flush_deferred_wgrads

                flush_deferred_wgrads()  # ZB: dW kernels under the recv wait
            grad = pipeline_communicate(
                "recv_backward", self.pp_rank, self.pp_size, self.group_ranks,
                shapes=tensor_shapes, dtype=dtype, device=device, group=self.pp_group,
            )
            b_in, b_out = inputs.pop(0), outputs.pop(0)
            if backwards_done + 1 == num_micro and not self._defer_wgrads:
                # (ZB: hooks stay off — post-accumulate hooks fire even
                # for deferred params, which would launch the bucket
                # reduction BEFORE the final flush; finalize launches)
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


class InterleavedOneFOneBSchedule(PipelineSchedule):
    """Interleaved (virtual-pipeline) 1F1B — Megatron-style, beyond
    reference parity.

    Each rank holds ``v`` model chunks (``InterleavedPipelineWrapper``);
    global stage ``c*p + r`` lives on rank r, so every inter-stage edge is
    rank r -> r+1 with a ring wrap between chunks.  Bubble fraction drops
    from (p-1)/(m+p-1) to ~(p-1)/(v*m).  Template (rank-independent):
    forward step k runs chunk ``(k % (p*v)) // p`` on micro-batch
    ``(k // (p*v))*p + k % p``; backward mirrors with chunks reversed.
    Identical templates on every rank make the per-edge message order of
    producer and consumer provably equal, so FIFO P2P matching is safe;
    the forward and backward directions use DEDICATED duplicate
    communicators (mesh "pp_fwd"/"pp_bwd") because at p=2 both directions
    share a rank pair.

    Requires ``num_micro % p == 0``.  DDP bucket overlap is disabled
    during the step (per-param hooks would over-count across the out-of-
    order chunk backwards and fire on partial sums); the full flat-buffer
    reduction happens in ``finalize_gradients``.
    """

    def train_step(self, data_loader, tensor_shapes, device, dtype) -> Dict[str, float]:
        from collections import deque

        from ...core.comm import ring_recv, ring_send

        owner = self._chunks_owner()
        v = int(owner.num_chunks)
        p, r = self.pp_size, self.pp_rank
        num_micro = data_loader.grad_acc_steps
        if p < 2:
            raise ValueError("interleaved schedule needs pp_size >= 2")
        if num_micro % p != 0:
            raise ValueError(
                f"interleaved 1F1B needs grad_acc_steps ({num_micro}) divisible "
                f"by pp_size ({p})"
            )
        fwd_group = getattr(self.trainer, "pp_fwd_group", None) or self.pp_group
        bwd_group = getattr(self.trainer, "pp_bwd_group", None) or self.pp_group

        pv = p * v
        total = num_micro * v
        warmup = min((p - r - 1) * 2 + (v - 1) * p, total)
        last_g = pv - 1

        metrics: Dict[str, float] = {}
        inputs = [deque() for _ in range(v)]
        outputs = [deque() for _ in range(v)]
        batch_cache: Dict[int, Any] = {}
        send_reqs: List[Any] = []
        send_keep: List[torch.Tensor] = []
        self._set_grad_sync(False)

        def forward(k: int) -> None:
            c = (k % pv) // p
            mu = (k // pv) * p + (k % p)
            g = c * p + r
            if c == 0:
                # every rank consumes the loader once per micro-batch, in
                # order, to stay lockstep; the batch is cached for the
                # label consumer (last global stage)
                batch_cache[mu] = next(data_loader)
                if r != 0 and r != p - 1:
                    batch_cache.pop(mu)  # middle ranks only keep lockstep
            if g == 0:
                x_data = self._stage_input(batch_cache[mu], device)
                if r != p - 1:
                    batch_cache.pop(mu)
                x_in = None
                out = owner.forward(x_data, chunk_id=c)
            else:
                x_in = ring_recv(
                    r, p, self.group_ranks, -1, tensor_shapes, dtype, device,
                    group=fwd_group, requires_grad=True,
                )
                out = owner.forward(x_in, chunk_id=c)
            if g == last_g:
                loss = self._loss_and_metrics(
                    out, batch_cache.pop(mu), device, metrics
                ) / num_micro
                inputs[c].append(x_in)
                outputs[c].append(loss)
            else:
                send_reqs.extend(
                    ring_send(out, r, p, self.group_ranks, +1, group=fwd_group)
                )
                inputs[c].append(x_in)
                outputs[c].append(out)

        def backward(k: int) -> None:
            c = v - 1 - (k % pv) // p
            g = c * p + r
            b_in = inputs[c].popleft()
            b_out = outputs[c].popleft()
            if g == last_g:
                grad = None
            else:
                if self._defer_wgrads:
                    from ...ops import flush_deferred_wgrads

                    flush_deferred_wgrads()  # ZB: dW under the ring recv
                grad = ring_recv(
                    r, p, self.group_ranks, +1, tensor_shapes, dtype, device,
                    group=bwd_group,
                )
            if self._defer_wgrads:
                from ...ops import defer_wgrads

                with defer_wgrads.scope():
                    in_grad = self.model.backward(b_in, b_out, grad)
            else:
                in_grad = self.model.backward(b_in, b_out, grad)
            if g != 0 and in_grad is not None:
                send_keep.append(in_grad)
                send_reqs.extend(
                    ring_send(in_grad, r, p, self.group_ranks, -1, group=bwd_group)
                )

        for k in range(warmup):
            forward(k)
        for i in range(total - warmup):
            forward(warmup + i)
            backward(i)
        for i in range(total - warmup, total):
            backward(i)
        for rq in send_reqs:
            rq.wait()

        self._set_grad_sync(True)
        self._optimizer_step()
        return self._final_metrics(metrics, num_micro)

    def _chunks_owner(self):
        m = self.model
        for _ in range(3):
            if hasattr(m, "num_chunks"):
                return m
            if hasattr(m, "module"):
                m = m.module
            else:
                break
        raise ValueError(
            "interleaved schedule requires the model to be (or wrap) an "
            "InterleavedPipelineWrapper"
        )


def get_schedule(name: str, trainer, task_type: str) -> PipelineSchedule:
    name = (name or "1f1b").lower()
    if name in ("1f1b", "one_f_one_b", "onefoneb"):
        return OneFOneBSchedule(trainer, task_type)
    if name in ("afab", "all_forward_all_backward", "gpipe"):
        return AllFwdAllBwdSchedule(trainer, task_type)
    if name in ("interleaved", "interleaved_1f1b", "vpp"):
        return InterleavedOneFOneBSchedule(trainer, task_type)
    raise ValueError(f"unknown schedule {name!r}")
