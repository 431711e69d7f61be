"""ZeRO-2 AdamW: gradients AND optimizer state sharded over the DP group.

Pairs with ``DataParallel(bucket_config=BucketConfig(grad_reduce_op=
"reduce_scatter"))``: each bucket's gradients arrive as this rank's 1/dp
chunk (one RCCL ``reduce_scatter_tensor`` per bucket, overlapped with
backward) — 2/3 of the gradient wire bytes of the all-reduce + shard
scheme of ZeRO-1, and reduced-gradient memory is 1/dp at rest.

MI355X-native layout: parameters are re-pointed into per-bucket flat
buffers mirroring the DDP grad-bucket layout (padded to dp_size), so

* the optimizer step is the fused AdamW HIP kernel over each bucket's
  own chunk (fp32 master/m/v exist only for the chunk);
* the post-step parameter all-gather is one RCCL
  ``all_gather_into_tensor`` per bucket (async, overlapped).

Completes the ZeRO family next to optim/zero.py (stage 1) and
parallel/zero3.py (stage 3).  The reference never finished any ZeRO
stage (optimizers/zero.py is a TODO stub).
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from ..ops import adamw_step_flat, l2_norm

__all__ = ["Zero2AdamW"]


class Zero2AdamW:
    def __init__(
        self,
        ddp,
        lr: float = 1e-3,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.01,
        max_grad_norm: Optional[float] = None,
    ):
        if not getattr(ddp, "_reduce_scatter", False):
            raise ValueError(
                "Zero2AdamW needs a DataParallel built with "
                'BucketConfig(grad_reduce_op="reduce_scatter")'
            )
        if len(ddp.flat_grads) != 1:
            raise ValueError("Zero2AdamW: params must share one dtype")
        self.ddp = ddp
        self.backend = ddp.backend
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.max_grad_norm = max_grad_norm
        self.step_count = 0
        self._clipped_this_step = False
        self.ws = ddp.world_size
        self.rank = self.backend.get_rank()

        dt = next(iter(ddp.flat_grads.keys()))
        dev = ddp.flat_grads[dt].device
        self.dtype = dt

        # per-bucket flat param buffers mirroring the grad-bucket layout;
        # params become views so the bucket all-gather updates them in place
        self.param_flats: List[torch.Tensor] = []
        self.masters: List[torch.Tensor] = []
        self.exp_avg: List[torch.Tensor] = []
        self.exp_avg_sq: List[torch.Tensor] = []
        self._chunks: List[slice] = []
        for b in ddp.buckets:
            n = b.flat.numel()
            chunk = n // self.ws
            pf = torch.zeros(n, dtype=dt, device=dev)
            off = 0
            for p in b.params:
                k = p.numel()
                pf[off : off + k].copy_(p.data.reshape(-1).to(dt))
                if p.dtype == dt:
                    p.data = pf[off : off + k].view_as(p.data)
                off += k
            s = slice(self.rank * chunk, (self.rank + 1) * chunk)
            self.param_flats.append(pf)
            self.masters.append(pf[s].detach().float().clone())
            self.exp_avg.append(torch.zeros_like(self.masters[-1]))
            self.exp_avg_sq.append(torch.zeros_like(self.masters[-1]))
            self._chunks.append(s)
        self.step_dev = (
            torch.zeros((), dtype=torch.int64, device=dev)
            if dev.type == "cuda" else None
        )

    # ------------------------------------------------------------------
    def clip_grad_norm_(self, max_norm: float) -> torch.Tensor:
        """TRUE global-norm clip: each rank's chunks are disjoint, so the
        DP sum of per-chunk squared norms IS the global squared norm
        (pure-DP meshes; compose TP/PP via the trainer-level clip)."""
        sq = None
        for b in self.ddp.buckets:
            n = l2_norm([b.own_chunk])
            sq = n * n if sq is None else sq + n * n
        sq = sq.float()
        if dist.is_initialized() and self.ws > 1:
            self.backend.all_reduce_tensor(sq)
        norm = sq.sqrt()
        scale = (max_norm / (norm + 1e-6)).clamp_(max=1.0)
        for b in self.ddp.buckets:
            b.own_chunk.mul_(scale.to(b.own_chunk.dtype))
        self._clipped_this_step = True
        return norm

    @torch.no_grad()
    def step(self) -> None:
        self.step_count += 1
        if self.step_dev is not None:
            self.step_dev += 1
        if self.max_grad_norm and not self._clipped_this_step:
            self.clip_grad_norm_(self.max_grad_norm)
        self._clipped_this_step = False
        works = []
        for b, pf, m, ea, es, s in zip(
            self.ddp.buckets, self.param_flats, self.masters, self.exp_avg,
            self.exp_avg_sq, self._chunks,
        ):
            shard = pf[s]
            adamw_step_flat(
                shard, m, b.own_chunk, ea, es, self.step_count, self.lr,
                self.beta1, self.beta2, self.eps, self.weight_decay,
                step_dev=self.step_dev,
            )
            works.append(
                self.backend.all_gather_into_tensor(pf, shard.contiguous(),
                                                    async_op=True)
            )
        for w in works:
            if w is not None:
                w.wait()

    def zero_grad(self, set_to_none: bool = False) -> None:
        self.ddp.zero_grad()

    def refresh_master_(self) -> None:
        for pf, m, s in zip(self.param_flats, self.masters, self._chunks):
            m.copy_(pf[s].float())

    @property
    def param_groups(self):
        return [{"params": [p for b in self.ddp.buckets for p in b.params],
                 "lr": self.lr}]

    def state_dict(self):
        return {
            "step": self.step_count,
            "masters": self.masters,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
            "lr": self.lr,
            "dp_rank": self.rank,
            "dp_size": self.ws,
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        if self.step_dev is not None:
            self.step_dev.fill_(sd["step"])
        for dst, src in zip(self.masters, sd["masters"]):
            dst.copy_(src)
        for dst, src in zip(self.exp_avg, sd["exp_avg"]):
            dst.copy_(src)
        for dst, src in zip(self.exp_avg_sq, sd["exp_avg_sq"]):
            dst.copy_(src)
        self.lr = sd.get("lr", self.lr)
