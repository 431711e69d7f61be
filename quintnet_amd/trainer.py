"""Generic Trainer: epoch loop + pipeline/plain dispatch + metrics.

Parity with reference trainer.py:66-363 (Trainer for the ViT
classification path).  The train loader's batch size is the MICRO-batch
size; one optimizer step consumes ``grad_acc_steps`` micro-batches
(matching the reference's PipelineDataLoader contract).
"""

from __future__ import annotations

import os
import time
from typing import Any, Dict, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from .parallel import DataParallel, PipelineDataLoader, PipelineTrainer

__all__ = ["Trainer"]


def _unwrap(model):
    """Peel DataParallel(.module) and PipelineParallelWrapper(.local_module)
    down to the stage/model that owns the canonical state-dict names."""
    m = model
    for _ in range(4):
        if hasattr(m, "module"):
            m = m.module
        elif hasattr(m, "local_module") and not isinstance(
            m.local_module, torch.nn.Sequential
        ):
            m = m.local_module
        else:
            break
    return m


class Trainer:
    def __init__(self, model, train_loader, val_loader, config: Dict[str, Any], pg_manager):
        self.model = model
        self.train_loader = train_loader
        self.val_loader = val_loader
        self.config = config
        self.pg = pg_manager
        self.device = pg_manager.device if pg_manager is not None else (
            torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
        )
        self.lr = float(config.get("learning_rate", 1e-3))
        self.num_epochs = int(config.get("num_epochs", 1))
        self.grad_acc_steps = int(config.get("grad_acc_steps", 1))
        self.max_grad_norm = config.get("max_grad_norm", 1.0)
        self.task_type = config.get("task_type", "classification")
        self.optimizer = torch.optim.Adam(self.model.parameters(), lr=self.lr)
        self.criterion = nn.CrossEntropyLoss()
        self.pp_size = pg_manager.pp_size if pg_manager is not None else 1
        self.is_pipeline = self.pp_size > 1
        self.pipeline_trainer: Optional[PipelineTrainer] = None
        self.tensor_shapes = None
        self._best_metric: Optional[float] = None
        self._es_best: Optional[float] = None
        self._es_bad = 0
        if self.is_pipeline:
            self._setup_pipeline()

    @property
    def _cp_group(self):
        if (
            self.config.get("context_parallel")
            and self.pg is not None
            and "cp" in getattr(self.pg, "mesh_name", ())
        ):
            return self.pg.get_group("cp")
        return None

    # ------------------------------------------------------------------
    def _setup_pipeline(self) -> None:
        micro_b = self._infer_micro_batch()
        inner = _unwrap(self.model)
        seq, hidden = self._infer_seq_hidden(inner)
        self.tensor_shapes = (micro_b, seq, hidden)
        groups = self.pg.get_all_groups()
        self.pipeline_trainer = PipelineTrainer(
            model=self.model,
            optimizer=self.optimizer,
            criterion=self.criterion,
            defer_wgrads=bool(self.config.get("defer_wgrads", False)),
            pp_rank=self.pg.pp_rank,
            pp_size=self.pg.pp_size,
            pp_group=self.pg.get_group("pp"),
            pp_group_ranks=self.pg.get_group_ranks("pp"),
            schedule=self.config.get("schedule", "1f1b"),
            task_type=self.task_type,
            max_grad_norm=self.max_grad_norm,
            pp_fwd_group=groups.get("pp_fwd"),
            pp_bwd_group=groups.get("pp_bwd"),
            cp_group=self._cp_group,
            tp_group=(
                self.pg.get_group("tp")
                if "tp" in self.pg.mesh_name and self.pg.tp_size > 1
                else None
            ),
        )

    def _infer_micro_batch(self) -> int:
        try:
            return int(self.train_loader.batch_size)
        except (AttributeError, TypeError):
            batch = next(iter(self.train_loader))
            if isinstance(batch, dict):
                key = "images" if "images" in batch else next(iter(batch))
                return len(batch[key])
            return len(batch[0])

    def _infer_seq_hidden(self, inner) -> tuple:
        seq = getattr(inner, "seq_len", None) or self.config.get("seq_len")
        hidden = getattr(inner, "hidden_dim", None) or self.config.get("hidden_dim")
        if seq is None or hidden is None:
            raise ValueError(
                "cannot infer pipeline tensor shapes; set config seq_len/hidden_dim"
            )
        return int(seq), int(hidden)

    @property
    def _dtype(self) -> torch.dtype:
        return next(self.model.parameters()).dtype

    # ------------------------------------------------------------------
    def _build_lr_schedule(self):
        """Optional step-based LR schedule (config: lr_schedule in
        constant|linear|cosine, warmup_steps, min_lr).  Built lazily at
        fit() time — subclasses replace self.optimizer in __init__."""
        kind = self.config.get("lr_schedule") or "constant"
        if kind == "constant" and not int(self.config.get("warmup_steps", 0)):
            return None
        from .optim import LRSchedule

        try:
            steps_per_epoch = max(
                1, len(self.train_loader) // max(self.grad_acc_steps, 1)
            )
        except TypeError:  # sized loaders only; fall back to config
            steps_per_epoch = int(self.config.get("steps_per_epoch", 1000))
        total = int(self.config.get(
            "total_steps", self.num_epochs * steps_per_epoch
        ))
        sched = LRSchedule(
            self.optimizer,
            base_lr=self.lr,
            total_steps=total,
            warmup_steps=int(self.config.get("warmup_steps", 0)),
            kind=kind,
            min_lr=float(self.config.get("min_lr", 0.0)),
        )
        # mid-training resume: the ZeRO optimizers checkpoint their own
        # step counter — restart the schedule from there
        sched._step = int(getattr(self.optimizer, "step_count", 0))
        return sched

    def _lr_step(self) -> None:
        sched = getattr(self, "lr_scheduler", None)
        if sched is not None:
            sched.step()

    def fit(self) -> Dict[str, float]:
        self.lr_scheduler = self._build_lr_schedule()
        if self.pipeline_trainer is not None:
            self.pipeline_trainer.lr_scheduler = self.lr_scheduler
        wd = None
        if self.config.get("watchdog_timeout_s"):
            from .utils.watchdog import Watchdog

            wd = Watchdog(
                float(self.config["watchdog_timeout_s"]),
                kill_on_hang=bool(self.config.get("watchdog_kill", False)),
            ).start()
            self._watchdog = wd
        try:
            return self._fit_inner(wd)
        finally:
            if wd:
                wd.stop()

    def _fit_inner(self, wd) -> Dict[str, float]:
        history = {}
        for epoch in range(self.num_epochs):
            t0 = time.time()
            train_metrics = self._train_epoch(epoch)
            val_metrics = self._validate_epoch() if self.val_loader is not None else {}
            dt = time.time() - t0
            metrics = self._broadcast_metrics(train_metrics, val_metrics)
            if self._is_rank0():
                msg = f"[epoch {epoch+1}/{self.num_epochs}] {dt:.1f}s"
                for k, v in metrics.items():
                    msg += f" {k}={v:.4f}"
                print(msg, flush=True)
                mf = self.config.get("metrics_file")
                if mf:
                    # machine-readable training record: one JSON line per
                    # epoch, append-mode (survives resume)
                    import json

                    lr_now = None
                    sched = getattr(self, "lr_scheduler", None)
                    if sched is not None:
                        lr_now = sched.lr_at(max(sched._step - 1, 0))
                    gnorm = getattr(self, "_last_grad_norm", None)
                    if gnorm is not None:
                        gnorm = round(float(gnorm), 6)
                    with open(mf, "a") as f:
                        f.write(json.dumps({
                            "epoch": epoch + 1, "seconds": round(dt, 2),
                            "lr": lr_now, "grad_norm": gnorm,
                            **{k: float(v) for k, v in metrics.items()},
                        }) + "\n")
            history = metrics
            if wd:
                wd.beat()
            every = int(self.config.get("save_every", 0))
            if every and (epoch + 1) % every == 0 and epoch + 1 < self.num_epochs:
                self._save_checkpoint()
            if self.config.get("save_best"):
                # keep a separate best-validation shard set alongside the
                # rolling checkpoint (lower is better: loss-like metrics)
                key = self.config.get("best_metric", "val_loss")
                cur = metrics.get(key)
                if cur is not None and (
                    self._best_metric is None or cur < self._best_metric
                ):
                    self._best_metric = float(cur)
                    name = self.config.get("checkpoint_name", "final_model")
                    self._save_checkpoint(name=f"{name}_best")
            patience = int(self.config.get("early_stop_patience", 0))
            if patience:
                key = self.config.get("best_metric", "val_loss")
                cur = metrics.get(key)
                if cur is not None:
                    if self._es_best is None or cur < self._es_best - float(
                        self.config.get("early_stop_min_delta", 0.0)
                    ):
                        self._es_best, self._es_bad = float(cur), 0
                    else:
                        self._es_bad += 1
                        if self._es_bad >= patience:
                            if self._is_rank0():
                                print(f"[early stop] {key} flat for "
                                      f"{patience} epochs", flush=True)
                            break
        self._save_checkpoint()
        return history

    def _is_rank0(self) -> bool:
        return not dist.is_initialized() or dist.get_rank() == 0

    # -- per-phase tracing (the reference left this as TODO
    # stubs; here HIP-event timing is wired into the real step loop) ----
    def _phase_timer(self):
        if not self.config.get("profile_phases"):
            return None
        from .utils.profiling import PhaseTimer

        return PhaseTimer()

    def _report_phases(self, pt, steps: int) -> None:
        if pt is None or not self._is_rank0():
            return
        tot = pt.summary()
        per = {k: v / max(steps, 1) for k, v in sorted(tot.items())}
        line = "  ".join(f"{k}={v:.2f}ms" for k, v in per.items())
        print(f"[phases/step] {line}", flush=True)

    def _broadcast_metrics(self, train: Dict[str, float], val: Dict[str, float]) -> Dict[str, float]:
        # world-uniform key schema (non-last PP stages have no metrics;
        # MAX all-reduce propagates the last stage's values — K17 pattern)
        base = ("loss", "ppl") if self.task_type == "clm" else ("loss", "accuracy")
        out = {f"train_{k}": float(train.get(k, 0.0)) for k in base}
        if self.val_loader is not None:
            out.update({f"val_{k}": float(val.get(k, 0.0)) for k in base})
        if not dist.is_initialized():
            return out
        keys = sorted(out)
        vals = torch.tensor([out[k] for k in keys], dtype=torch.float64)
        if self.device.type == "cuda":
            vals = vals.to(self.device)
        dist.all_reduce(vals, op=dist.ReduceOp.MAX)
        return dict(zip(keys, vals.cpu().tolist()))

    # ------------------------------------------------------------------
    def _train_epoch(self, epoch: int) -> Dict[str, float]:
        self.model.train()
        if self.is_pipeline:
            return self._train_epoch_pipeline()
        return self._train_epoch_plain()

    def _train_epoch_pipeline(self) -> Dict[str, float]:
        from .utils.profiling import StepTimer

        loader = PipelineDataLoader(self.train_loader, self.grad_acc_steps, self.task_type)
        num_steps = max(len(self.train_loader) // self.grad_acc_steps, 1)
        agg: Dict[str, float] = {}
        timer = StepTimer() if self.config.get("profile") else None
        pt = self._phase_timer()
        self.pipeline_trainer.phase_timer = pt
        wd = getattr(self, "_watchdog", None)
        for _ in range(num_steps):
            if wd:
                wd.beat()
            if timer:
                timer.start()
            m = self.pipeline_trainer.train_step(
                loader, self.tensor_shapes, self.device, self._dtype
            )
            if timer:
                timer.stop()
            for k, v in m.items():
                agg[k] = agg.get(k, 0.0) + float(v)
        out = {k: v / num_steps for k, v in agg.items()}
        self._report_phases(pt, num_steps)
        if timer and self._is_rank0():
            bs = self._infer_micro_batch() * self.grad_acc_steps
            print(
                f"[profile] {timer.mean_ms:.1f} ms/step, "
                f"{1000.0 * bs / max(timer.mean_ms, 1e-9):.1f} samples/s/replica",
                flush=True,
            )
        return out

    def _train_epoch_plain(self) -> Dict[str, float]:
        total_loss, correct, total, steps = 0.0, 0, 0, 0
        accum = 0
        is_ddp = isinstance(self.model, DataParallel)
        for batch in self.train_loader:
            batch = PipelineDataLoader._normalize(batch)
            x = batch["images"].to(self.device, non_blocking=True)
            y = batch["labels"].to(self.device, non_blocking=True)
            if is_ddp:
                # async bucket reduce only on the LAST micro-batch of the
                # window (same rule as gpt2_trainer.py — earlier micro-
                # batches would race the reduce with accumulation)
                self.model.require_backward_grad_sync = (
                    accum + 1 == self.grad_acc_steps
                )
            out = self.model(x)
            loss = self.criterion(out, y)
            (loss / self.grad_acc_steps).backward()
            accum += 1
            total_loss += float(loss.detach())
            correct += int((out.argmax(-1) == y).sum())
            total += int(y.numel())
            steps += 1
            if accum < self.grad_acc_steps:
                continue
            accum = 0
            if is_ddp:
                self.model.finalize_gradients()
            if self.max_grad_norm:
                from .ops import clip_grad_norm_global

                tp_group = (
                    self.pg.get_group("tp")
                    if self.pg is not None
                    and "tp" in getattr(self.pg, "mesh_name", ())
                    and self.pg.tp_size > 1
                    else None
                )
                clip_grad_norm_global(
                    [p for p in self.model.parameters() if p.requires_grad],
                    self.max_grad_norm,
                    tp_group=tp_group,
                )
            self._lr_step()
            self.optimizer.step()
            if hasattr(self.model, "zero_grad") and isinstance(self.model, DataParallel):
                self.model.zero_grad()
            else:
                self.optimizer.zero_grad()
        if accum:
            # tail window (loader length not a multiple of grad_acc_steps)
            if is_ddp:
                self.model.require_backward_grad_sync = True
                self.model.finalize_gradients()
            self._lr_step()
            self.optimizer.step()
            if is_ddp:
                self.model.zero_grad()
            else:
                self.optimizer.zero_grad()
        return {
            "loss": total_loss / max(steps, 1),
            "accuracy": 100.0 * correct / max(total, 1),
        }

    # ------------------------------------------------------------------
    @torch.no_grad()
    def _validate_epoch(self) -> Dict[str, float]:
        self.model.eval()
        if self.is_pipeline:
            return self.pipeline_trainer.evaluate(
                self.val_loader, self.tensor_shapes, self.device, self._dtype
            )
        total_loss, correct, total, steps = 0.0, 0, 0, 0
        for batch in self.val_loader:
            batch = PipelineDataLoader._normalize(batch)
            x = batch["images"].to(self.device, non_blocking=True)
            y = batch["labels"].to(self.device, non_blocking=True)
            out = self.model(x)
            total_loss += float(self.criterion(out, y))
            correct += int((out.argmax(-1) == y).sum())
            total += y.numel()
            steps += 1
        metrics = {
            "loss": total_loss / max(steps, 1),
            "accuracy": 100.0 * correct / max(total, 1),
        }
        if dist.is_initialized() and self.pg is not None and self.pg.dp_size > 1:
            t = torch.tensor([metrics["loss"], metrics["accuracy"]], dtype=torch.float64)
            if self.device.type == "cuda":
                t = t.to(self.device)
            dist.all_reduce(t, op=dist.ReduceOp.SUM, group=self.pg.get_group("dp"))
            t /= self.pg.dp_size
            metrics = {"loss": float(t[0]), "accuracy": float(t[1])}
        return metrics

    # ------------------------------------------------------------------
    def _save_checkpoint(self, name: str = None) -> None:
        path = self.config.get("checkpoint_dir")
        if not path:
            return
        os.makedirs(path, exist_ok=True)
        if self._is_rank0():
            torch.save(
                {"model_state_dict": _unwrap(self.model).state_dict(), "config": self.config},
                os.path.join(path, f"{name}.pt" if name else "checkpoint.pt"),
            )
