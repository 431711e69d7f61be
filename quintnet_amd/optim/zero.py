"""ZeRO-1 AdamW: optimizer state sharded across the DP group.

The reference never finished this (optimizers/zero.py is a TODO stub);
BASELINE.json's GPT-2 config requires it.  MI355X-native design:

* all trainable params are re-pointed into ONE flat contiguous buffer
  (model dtype, padded to dp_size), so the post-step parameter
  all-gather is a single RCCL ``all_gather_into_tensor`` on the flat
  buffer — no per-param traffic;
* gradients live in ONE matching flat buffer.  Three modes:
  - shared: reuse the DataParallel wrapper's bucket buffer
    (``ZeroRedundancyAdamW.from_ddp``) — zero copies, grads arrive
    already averaged by the bucketed RCCL all-reduce;
  - owned: no DDP — the optimizer allocates the buffer and re-points
    ``p.grad`` into it, so autograd accumulates in place;
  - copy: fallback when grads are externally managed.
* each DP rank owns a 1/dp_size shard and keeps ONLY its shard's fp32
  master weights + Adam m/v (the ZeRO-1 memory win);
* the update is the fused AdamW HIP kernel (csrc/adamw.hip) over the
  flat shard; grad-norm clipping is two kernels on the flat buffer.
"""

from __future__ import annotations

from typing import Iterable, List, Optional

import torch
import torch.distributed as dist

from ..ops import adamw_step_flat, l2_norm

__all__ = ["ZeroRedundancyAdamW", "DistributedAdamW"]


class ZeroRedundancyAdamW:
    def __init__(
        self,
        params: Iterable[torch.nn.Parameter],
        lr: float = 1e-3,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.01,
        dp_group=None,
        max_grad_norm: Optional[float] = None,
        grad_buffer: Optional[torch.Tensor] = None,
        tp_group=None,
        pp_group=None,
    ):
        self.params: List[torch.nn.Parameter] = [p for p in params if p.requires_grad]
        if not self.params:
            raise ValueError("no trainable parameters")
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.dp_group = dp_group
        self.max_grad_norm = max_grad_norm
        self.tp_group = tp_group
        self.pp_group = pp_group
        self.step_count = 0
        self._clipped_this_step = False  # guards against double clipping

        # dp_group=None means NO sharding (a PP/TP-only rank must not
        # shard over the world group — ranks hold different params).
        # Pass dist.group.WORLD explicitly for world-wide ZeRO.
        if dp_group is None or not dist.is_initialized():
            self.dp_size, self.dp_rank = 1, 0
            self.dp_group = None
        else:
            self.dp_size = dist.get_world_size(group=dp_group)
            self.dp_rank = dist.get_rank(group=dp_group)

        device = self.params[0].device
        self.dtype = self.params[0].dtype
        total = sum(p.numel() for p in self.params)
        self.padded = ((total + self.dp_size - 1) // self.dp_size) * self.dp_size
        self.shard_size = self.padded // self.dp_size

        uniform_dtype = all(p.dtype == self.dtype for p in self.params)

        # one flat param buffer; params become views into it
        self.flat_param = torch.zeros(self.padded, dtype=self.dtype, device=device)
        off = 0
        self._offsets: List[int] = []
        for p in self.params:
            n = p.numel()
            self.flat_param[off : off + n].copy_(p.data.reshape(-1).to(self.dtype))
            if p.dtype == self.dtype:
                p.data = self.flat_param[off : off + n].view_as(p.data)
            self._offsets.append(off)
            off += n

        # grad buffer: shared (DDP) / owned / copy
        if grad_buffer is not None:
            assert grad_buffer.numel() >= self.padded, "shared grad buffer too small"
            self.flat_grad = grad_buffer
            self._grad_mode = "shared"
        elif uniform_dtype and all(p.grad is None for p in self.params):
            self.flat_grad = torch.zeros(self.padded, dtype=self.dtype, device=device)
            off = 0
            for p in self.params:
                p.grad = self.flat_grad[off : off + p.numel()].view_as(p.data)
                off += p.numel()
            self._grad_mode = "owned"
        else:
            self.flat_grad = torch.zeros(self.padded, dtype=self.dtype, device=device)
            self._grad_mode = "copy"

        s = self.dp_rank * self.shard_size
        e = s + self.shard_size
        self._shard_slice = slice(s, e)
        self.master = self.flat_param[s:e].detach().float().clone()
        self.exp_avg = torch.zeros_like(self.master)
        self.exp_avg_sq = torch.zeros_like(self.master)
        # device-side step counter: advanced ON DEVICE inside step() so
        # bias correction stays correct under hipGraph replay
        self.step_dev = (
            torch.zeros((), dtype=torch.int64, device=device) if device.type == "cuda" else None
        )

    @classmethod
    def from_ddp(cls, ddp, **kw):
        """Build over a DataParallel wrapper sharing its grad buffer
        (bucket order == flat order, zero gather copies)."""
        buf = ddp.grad_buffer()
        params = ddp.grad_buffer_params()
        if buf is None or params is None:
            return cls(ddp.parameters(), **kw)
        return cls(params, grad_buffer=buf, **kw)

    # ------------------------------------------------------------------
    def _gather_grads(self) -> None:
        if self._grad_mode != "copy":
            return
        self.flat_grad.zero_()
        for p, off in zip(self.params, self._offsets):
            if p.grad is not None:
                self.flat_grad[off : off + p.numel()].copy_(p.grad.reshape(-1))

    def clip_grad_norm_(self, max_norm: float) -> torch.Tensor:
        """Global-norm clip on the flat grad buffer.  The norm counts
        each logical parameter once across TP/PP (sharded params summed
        over tp_group, per-stage params over pp_group, tied LM-head copy
        skipped — see ops.grad_sq_norm_contrib); the DP axis needs no
        reduction because grads arrive already averaged.  Fully
        device-side (no host sync — hipGraph-capture safe): always
        multiplies by min(max/norm, 1)."""
        import torch.distributed as _d

        from ..ops.adamw import _group_active

        self._gather_grads()
        tp_on = _group_active(self.tp_group)
        pp_on = _group_active(self.pp_group)
        if tp_on or pp_on:
            from ..ops import grad_sq_norm_contrib

            tp_rank = _d.get_rank(group=self.tp_group) if tp_on else 0
            sq, _ = grad_sq_norm_contrib(self.params, tp_rank)
            if sq is None:
                sq = torch.zeros((), dtype=torch.float32,
                                 device=self.flat_grad.device)
            if tp_on:
                _d.all_reduce(sq, op=_d.ReduceOp.SUM, group=self.tp_group)
            if pp_on:
                _d.all_reduce(sq, op=_d.ReduceOp.SUM, group=self.pp_group)
            norm = sq.sqrt()
        else:
            norm = l2_norm([self.flat_grad])
        scale = (max_norm / (norm + 1e-6)).clamp_(max=1.0)
        self.flat_grad.mul_(scale.to(self.flat_grad.dtype))
        self._clipped_this_step = True
        return norm

    @torch.no_grad()
    def step(self) -> None:
        self.step_count += 1
        if self.step_dev is not None:
            self.step_dev += 1  # device op: correct under graph replay
        self._gather_grads()
        if self.max_grad_norm and not self._clipped_this_step:
            # skip when the schedule already clipped this step (the
            # scale would otherwise be applied twice)
            self.clip_grad_norm_(self.max_grad_norm)
        self._clipped_this_step = False
        shard_param = self.flat_param[self._shard_slice]
        shard_grad = self.flat_grad[self._shard_slice]
        adamw_step_flat(
            shard_param,
            self.master,
            shard_grad,
            self.exp_avg,
            self.exp_avg_sq,
            self.step_count,
            self.lr,
            self.beta1,
            self.beta2,
            self.eps,
            self.weight_decay,
            step_dev=self.step_dev,
        )
        if self.dp_size > 1:
            dist.all_gather_into_tensor(
                self.flat_param, shard_param.contiguous(), group=self.dp_group
            )
        # params whose dtype differs from the flat buffer keep their own
        # storage — copy their slice back
        for p, off in zip(self.params, self._offsets):
            if p.dtype != self.dtype:
                p.data.copy_(self.flat_param[off : off + p.numel()].view_as(p).to(p.dtype))

    def refresh_master_(self) -> None:
        """Re-sync the fp32 master shard from the (possibly just-loaded)
        model params.  Call after load_state_dict on the MODEL when no
        optimizer checkpoint is restored."""
        self.master.copy_(self.flat_param[self._shard_slice].float())

    def zero_grad(self, set_to_none: bool = False) -> None:
        if self._grad_mode == "owned":
            self.flat_grad.zero_()
            return
        if self._grad_mode == "shared":
            # the DataParallel wrapper zeroes its own buffer
            self.flat_grad.zero_()
            return
        for p in self.params:
            if p.grad is not None:
                if set_to_none:
                    p.grad = None
                else:
                    p.grad.zero_()

    # minimal torch-optimizer-compatible surface
    @property
    def param_groups(self):
        return [{"params": self.params, "lr": self.lr}]

    def state_dict(self):
        return {
            "step": self.step_count,
            "master": self.master,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
            "lr": self.lr,
            "dp_rank": self.dp_rank,
            "dp_size": self.dp_size,
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        if self.step_dev is not None:
            self.step_dev.fill_(sd["step"])
        self.master.copy_(sd["master"])
        self.exp_avg.copy_(sd["exp_avg"])
        self.exp_avg_sq.copy_(sd["exp_avg_sq"])
        self.lr = sd.get("lr", self.lr)


# Unsharded fused distributed AdamW (dp_size==1 degenerate of ZeRO-1)
class DistributedAdamW(ZeroRedundancyAdamW):
    def __init__(self, params, **kw):
        kw.setdefault("dp_group", None)
        super().__init__(params, **kw)
