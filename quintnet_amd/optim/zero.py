"""ZeRO-1 AdamW: optimizer state sharded across the DP group.

The reference never finished this (optimizers/zero.py is a TODO stub);
BASELINE.json's GPT-2 config requires it.  MI355X-native design:

* all trainable params are re-pointed into ONE flat contiguous buffer
  (model dtype, padded to dp_size), so the post-step parameter
  all-gather is a single RCCL ``all_gather_into_tensor`` on the flat
  buffer — no per-param traffic;
* each DP rank owns a 1/dp_size shard and keeps ONLY its shard's fp32
  master weights + Adam m/v (the ZeRO-1 memory win);
* the update itself is the fused multi-tensor AdamW HIP kernel
  (csrc/adamw.hip) over the flat shard;
* gradients arrive already averaged by DataParallel's bucketed
  all-reduce; ``step()`` copies them into the flat grad layout, updates
  the owned shard, and all-gathers the new params.
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
        self.step_count = 0

        self.dp_size = (
            dist.get_world_size(group=dp_group) if dist.is_initialized() else 1
        )
        self.dp_rank = dist.get_rank(group=dp_group) if dist.is_initialized() else 0

        device = self.params[0].device
        self.dtype = self.params[0].dtype
        total = sum(p.numel() for p in self.params)
        self.padded = ((total + self.dp_size - 1) // self.dp_size) * self.dp_size
        self.shard_size = self.padded // self.dp_size

        # one flat param buffer; params become views into it
        self.flat_param = torch.zeros(self.padded, dtype=self.dtype, device=device)
        off = 0
        self._offsets: List[int] = []
        for p in self.params:
            n = p.numel()
            self.flat_param[off : off + n].copy_(p.data.reshape(-1))
            p.data = self.flat_param[off : off + n].view_as(p.data)
            self._offsets.append(off)
            off += n
        self.flat_grad = torch.zeros_like(self.flat_param)

        s = self.dp_rank * self.shard_size
        e = s + self.shard_size
        self._shard_slice = slice(s, e)
        self.master = self.flat_param[s:e].detach().float().clone()
        self.exp_avg = torch.zeros_like(self.master)
        self.exp_avg_sq = torch.zeros_like(self.master)

    # ------------------------------------------------------------------
    def _gather_grads(self) -> None:
        self.flat_grad.zero_()
        for p, off in zip(self.params, self._offsets):
            if p.grad is not None:
                self.flat_grad[off : off + p.numel()].copy_(p.grad.reshape(-1))

    def grad_global_norm(self) -> torch.Tensor:
        """Global L2 grad norm (exact across TP/PP when those grads are
        local — caller reduces if needed)."""
        return l2_norm([p.grad for p in self.params])

    @torch.no_grad()
    def step(self) -> None:
        self.step_count += 1
        self._gather_grads()
        if self.max_grad_norm:
            norm = l2_norm([self.flat_grad])
            scale = self.max_grad_norm / (float(norm) + 1e-6)
            if scale < 1.0:
                self.flat_grad.mul_(scale)
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
        )
        if self.dp_size > 1:
            dist.all_gather_into_tensor(
                self.flat_param, shard_param.contiguous(), group=self.dp_group
            )

    def zero_grad(self, set_to_none: bool = False) -> None:
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
        self.master.copy_(sd["master"])
        self.exp_avg.copy_(sd["exp_avg"])
        self.exp_avg_sq.copy_(sd["exp_avg_sq"])
        self.lr = sd.get("lr", self.lr)


# Unsharded fused distributed AdamW (dp_size==1 degenerate of ZeRO-1)
class DistributedAdamW(ZeroRedundancyAdamW):
    def __init__(self, params, **kw):
        kw.setdefault("dp_group", None)
        super().__init__(params, **kw)
