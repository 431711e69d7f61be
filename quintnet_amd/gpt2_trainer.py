"""GPT2Trainer: causal-LM training with ZeRO-1 AdamW + shard checkpoints.

Parity with reference GPT2_Trainer.py:67-555 (AdamW wd=0.01, CE with
ignore_index=-100, perplexity metrics, tied-weight grad sync each step,
per-rank shard checkpoints ``{name}_pp{p}_tp{t}.pt`` with parallelism
metadata).  The optimizer is the ZeRO-1 sharded AdamW (quintnet_amd.optim)
— the reference's unfinished design, completed here.
"""

from __future__ import annotations

import math
import os
from typing import Any, Dict

import torch
import torch.distributed as dist
import torch.nn as nn

from .optim import ZeroRedundancyAdamW
from .parallel import DataParallel
from .trainer import Trainer, _unwrap

__all__ = ["GPT2Trainer"]


class GPT2Trainer(Trainer):
    def __init__(self, model, train_loader, val_loader, config: Dict[str, Any], pg_manager):
        config = dict(config)
        config.setdefault("task_type", "clm")
        super().__init__(model, train_loader, val_loader, config, pg_manager)
        wd = float(config.get("weight_decay", 0.01))
        use_zero = bool(config.get("zero1", True))
        dp_group = (
            pg_manager.get_group("dp")
            if pg_manager is not None and "dp" in pg_manager.mesh_name and pg_manager.dp_size > 1
            else None
        )
        def _axis_group(axis, size):
            if pg_manager is None or axis not in getattr(pg_manager, "mesh_name", ()):
                return None
            return pg_manager.get_group(axis) if size > 1 else None

        tp_group = _axis_group("tp", pg_manager.tp_size if pg_manager else 1)
        pp_group = _axis_group("pp", pg_manager.pp_size if pg_manager else 1)
        self._tp_group, self._pp_group = tp_group, pp_group
        if use_zero:
            zkw = dict(lr=self.lr, weight_decay=wd, dp_group=dp_group,
                       max_grad_norm=None,  # clipping handled by the schedule
                       tp_group=tp_group, pp_group=pp_group)
            if isinstance(self.model, DataParallel) and getattr(
                self.model, "_reduce_scatter", False
            ):
                # zero_stage 2 (config → coordinator built reduce-scatter
                # buckets): grads AND optimizer state sharded per bucket
                from .optim import Zero2AdamW

                self.optimizer = Zero2AdamW(
                    self.model, lr=self.lr, weight_decay=wd, max_grad_norm=None
                )
            elif isinstance(self.model, DataParallel):
                self.optimizer = ZeroRedundancyAdamW.from_ddp(self.model, **zkw)
            else:
                self.optimizer = ZeroRedundancyAdamW(self.model.parameters(), **zkw)
        else:
            self.optimizer = torch.optim.AdamW(
                self.model.parameters(), lr=self.lr, weight_decay=wd
            )
        self._label_smoothing = float(config.get("label_smoothing", 0.0))
        self.criterion = nn.CrossEntropyLoss(
            ignore_index=-100, label_smoothing=self._label_smoothing
        )
        if self.is_pipeline:
            # re-bind the pipeline trainer to the new optimizer/criterion
            self.pipeline_trainer.optimizer = self.optimizer
            self.pipeline_trainer.criterion = self.criterion
        resume = config.get("resume_from")
        if resume == "auto":
            # elastic restart convenience: resume from checkpoint_dir when
            # its shards exist, start fresh otherwise
            cdir = config.get("checkpoint_dir") or config.get("output_dir")
            name = config.get("checkpoint_name", "final_model")
            probe = os.path.join(cdir or "", f"{name}_pp0_tp0.pt")
            resume = cdir if cdir and os.path.exists(probe) else None
        if resume:
            from .checkpoint import load_sharded_checkpoint

            load_sharded_checkpoint(
                _unwrap(self.model), resume,
                name=config.get("checkpoint_name", "final_model"),
                pg_manager=pg_manager, optimizer=self.optimizer,
            )

    # ------------------------------------------------------------------
    def _infer_seq_hidden(self, inner) -> tuple:
        mc = self.config.get("model_config", {})
        # prefer explicit config; fall back to the stage's own hints before
        # assuming GPT-2-small defaults
        seq = self.config.get("max_seq_length") or mc.get("n_positions") or getattr(
            inner, "seq_len", None
        ) or 1024
        hidden = mc.get("n_embd") or getattr(inner, "hidden_dim", None) or 768
        seq, hidden = int(seq), int(hidden)
        if mc.get("sequence_parallel") and self.pg is not None and self.pg.tp_size > 1:
            seq //= self.pg.tp_size  # inter-stage activations are seq shards
        if self._cp_group is not None:
            seq //= self.pg.axis_size("cp")  # CP: pipeline carries shards
        return seq, hidden

    # ------------------------------------------------------------------
    def _train_epoch_plain(self) -> Dict[str, float]:
        total_loss, total_tokens, steps = 0.0, 0, 0
        accum = 0
        from .ops import causal_lm_loss

        cp_group = self._cp_group
        pt = self._phase_timer()
        is_ddp = isinstance(self.model, DataParallel)
        for batch in self.train_loader:
            ids = batch["input_ids"].to(self.device, non_blocking=True)
            labels = batch["labels"].to(self.device, non_blocking=True)
            if is_ddp:
                # bucket hooks may only fire the async all-reduce on the
                # LAST micro-batch of the accumulation window; earlier
                # micro-batches would race the in-flight reduce with the
                # still-accumulating grads.
                self.model.require_backward_grad_sync = (
                    accum + 1 == self.grad_acc_steps
                )
            if pt:
                pt.start("forward")
            if cp_group is not None:
                from .parallel import (
                    cp_causal_lm_loss,
                    scatter_clm_targets,
                    scatter_to_context,
                    zigzag_clm_targets,
                    zigzag_to_context,
                )

                # shard layout must match the stage's position math:
                # zigzag configs expect chunks (r, 2cp-1-r), contiguous
                # configs expect rank-ordered slices
                inner = _unwrap(self.model)
                if getattr(getattr(inner, "config", None), "cp_zigzag", False):
                    ids_shard = zigzag_to_context(ids, cp_group, dim=1)
                    tgt_shard = zigzag_clm_targets(labels, cp_group)
                else:
                    ids_shard = scatter_to_context(ids, cp_group, dim=1)
                    tgt_shard = scatter_clm_targets(labels, cp_group)
                logits = self.model(ids_shard)
                loss, true_loss = cp_causal_lm_loss(logits, tgt_shard, cp_group)
            else:
                logits = self.model(ids)
                loss = causal_lm_loss(logits, labels, ignore_index=-100,
                                      label_smoothing=self._label_smoothing)
            aux_w = float(self.config.get("moe_aux_weight", 0.0))
            total = loss
            if aux_w:
                inner = _unwrap(self.model)
                if hasattr(inner, "moe_aux_loss"):
                    total = loss + aux_w * inner.moe_aux_loss()
            if pt:
                pt.stop("forward")
                pt.start("backward")
            (total / self.grad_acc_steps).backward()
            if pt:
                pt.stop("backward")
            accum += 1
            total_loss += float(true_loss if cp_group is not None else loss.detach())
            total_tokens += int((labels[:, 1:] != -100).sum())
            steps += 1
            if accum == self.grad_acc_steps:
                accum = 0
                self._optim_step(pt)
        if accum:
            # tail window (loader length not a multiple of grad_acc_steps):
            # hooks never fired (sync was off), so enable sync and let
            # finalize_gradients launch every bucket itself.
            if is_ddp:
                self.model.require_backward_grad_sync = True
            self._optim_step(pt)
        self._report_phases(pt, steps)
        avg = total_loss / max(steps, 1)
        return {"loss": avg, "ppl": math.exp(min(avg, 20.0)), "n_tokens": total_tokens}

    def _optim_step(self, pt) -> None:
        """finalize grads → tied/SP sync → [NaN guard] → clip → step."""
        if pt:
            pt.start("grad_comm")
        if isinstance(self.model, DataParallel):
            self.model.finalize_gradients()
        inner = _unwrap(self.model)
        if hasattr(inner, "sync_tied_weights_grad"):
            inner.sync_tied_weights_grad()
        if pt:
            pt.stop("grad_comm")
            pt.start("optimizer")
        if self.config.get("detect_nan_grads"):
            from .utils.watchdog import assert_finite_grads

            assert_finite_grads(self.model)
        if self.max_grad_norm:
            if hasattr(self.optimizer, "clip_grad_norm_"):
                gn = self.optimizer.clip_grad_norm_(self.max_grad_norm)
            else:
                from .ops import clip_grad_norm_global

                gn = clip_grad_norm_global(
                    [p for p in self.model.parameters() if p.requires_grad],
                    self.max_grad_norm,
                    tp_group=self._tp_group,
                    pp_group=self._pp_group,
                )
            # observability: last pre-clip global grad norm (float lazily
            # — avoid a sync inside the hot loop; .item() on read)
            self._last_grad_norm = gn
        self._lr_step()
        self.optimizer.step()
        if isinstance(self.model, DataParallel):
            self.model.zero_grad()
        else:
            self.optimizer.zero_grad()
        if pt:
            pt.stop("optimizer")

    @torch.no_grad()
    def _validate_epoch(self) -> Dict[str, float]:
        self.model.eval()
        if self.is_pipeline:
            return self.pipeline_trainer.evaluate(
                self.val_loader, self.tensor_shapes, self.device, self._dtype
            )
        from .ops import causal_lm_loss

        total_loss, steps = 0.0, 0
        for batch in self.val_loader:
            ids = batch["input_ids"].to(self.device, non_blocking=True)
            labels = batch["labels"].to(self.device, non_blocking=True)
            logits = self.model(ids)
            total_loss += float(causal_lm_loss(logits, labels, ignore_index=-100))
            steps += 1
        avg = total_loss / max(steps, 1)
        return {"loss": avg, "ppl": math.exp(min(avg, 20.0))}

    # ------------------------------------------------------------------
    def _save_checkpoint(self, name: str = None) -> None:
        out_dir = self.config.get("checkpoint_dir") or self.config.get("output_dir")
        if not out_dir:
            return
        from .checkpoint import save_sharded_checkpoint

        name = name or self.config.get("checkpoint_name", "final_model")
        save_sharded_checkpoint(
            _unwrap(self.model),
            out_dir,
            name=name,
            pg_manager=self.pg,
            optimizer=self.optimizer,
            config=self.config,
        )

    # generation-quality metrics hook (reference GPT2_Trainer.py:509-555)
    @torch.no_grad()
    def evaluate_generation(self, tokenizer, prompts, max_new_tokens: int = 64):
        from .utils.metrics import generate_greedy

        self.model.eval()
        outs = []
        for p in prompts:
            outs.append(
                generate_greedy(self.model, tokenizer, p, max_new_tokens, self.device)
            )
        return outs
