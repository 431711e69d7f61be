"""Expert parallelism (MoE), beyond reference parity.

``ExpertParallelMLP`` is a drop-in MLP replacement: a top-k softmax
router sends each token to its experts; experts are sharded across the
EP group and tokens travel by a VARIABLE-SIZE all-to-all.  The exchange
is one custom autograd Function built on batched ``isend/irecv`` pairs
(exact per-destination counts — no capacity factor, no dropped tokens,
no padding), so it runs identically over gloo (tests) and RCCL/xGMI
(one EP hop per MoE layer each way).  Backward reverses the exchange
with transposed split sizes.

The switch-style load-balancing auxiliary loss is returned alongside
the output; add ``aux_weight * aux`` to the training loss.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

__all__ = ["ExpertParallelMLP", "all_to_all_var"]


def _ws(group) -> int:
    if group is None or not dist.is_initialized():
        return 1
    return dist.get_world_size(group=group)


class _AllToAllVar(torch.autograd.Function):
    """Exchange row-blocks of x: rank r sends x[splits before i] to rank i.

    ``out_splits[i]`` rows go TO rank i; ``in_splits[i]`` rows arrive FROM
    rank i.  Deterministic batched P2P; backward swaps the split lists.
    """

    @staticmethod
    def forward(ctx, x, out_splits, in_splits, group):
        ctx.group = group
        ctx.out_splits = out_splits
        ctx.in_splits = in_splits
        return _exchange(x, out_splits, in_splits, group)

    @staticmethod
    def backward(ctx, grad):
        return (
            _exchange(grad.contiguous(), ctx.in_splits, ctx.out_splits, ctx.group),
            None, None, None,
        )


def _exchange(x, out_splits, in_splits, group):
    world = _ws(group)
    if world == 1:
        return x
    rank = dist.get_rank(group=group)
    ranks = dist.get_process_group_ranks(group)
    chunks = list(x.split(out_splits, dim=0))
    recv = [
        torch.empty(n, *x.shape[1:], dtype=x.dtype, device=x.device)
        for n in in_splits
    ]
    ops = []
    for peer in range(world):
        if peer == rank:
            continue
        if out_splits[peer] > 0:
            ops.append(dist.P2POp(dist.isend, chunks[peer].contiguous(),
                                  peer=ranks[peer], group=group))
        if in_splits[peer] > 0:
            ops.append(dist.P2POp(dist.irecv, recv[peer],
                                  peer=ranks[peer], group=group))
    if ops:
        for r in dist.batch_isend_irecv(ops):
            r.wait()
    recv[rank] = chunks[rank]
    return torch.cat(recv, dim=0)


def all_to_all_var(x, out_splits, in_splits, group):
    return _AllToAllVar.apply(x, out_splits, in_splits, group)


class _Expert(nn.Module):
    """One expert MLP; with a tp_group its two GEMMs are Megatron
    column/row-parallel (TP peers must hold IDENTICAL tokens — routing is
    deterministic on replicated router weights, so they do)."""

    def __init__(self, n_embd: int, n_inner: int, device=None, dtype=None,
                 tp_group=None):
        super().__init__()
        kw = {"device": device, "dtype": dtype}
        if tp_group is not None:
            from .tensor_parallel import ColumnParallelLinear, RowParallelLinear

            self.fc1 = ColumnParallelLinear(
                n_embd, n_inner, tp_group=tp_group, gather_output=False, **kw
            )
            self.fc2 = RowParallelLinear(
                n_inner, n_embd, tp_group=tp_group, input_is_parallel=True, **kw
            )
        else:
            self.fc1 = nn.Linear(n_embd, n_inner, **kw)
            self.fc2 = nn.Linear(n_inner, n_embd, **kw)

    def forward(self, x):
        return self.fc2(torch.nn.functional.gelu(self.fc1(x), approximate="tanh"))


class ExpertParallelMLP(nn.Module):
    def __init__(
        self,
        n_embd: int,
        n_inner: int,
        n_experts: int,
        top_k: int = 2,
        ep_group=None,
        tp_group=None,
        device=None,
        dtype=None,
        capacity_factor: float = 0.0,
    ):
        super().__init__()
        self.ep_group = ep_group
        self.ep_size = _ws(ep_group)
        self.ep_rank = dist.get_rank(group=ep_group) if self.ep_size > 1 else 0
        if n_experts % self.ep_size != 0:
            raise ValueError("n_experts must divide by the EP group size")
        self.n_experts = n_experts
        self.n_local = n_experts // self.ep_size
        self.top_k = top_k
        kw = {"device": device, "dtype": dtype}
        self.router = nn.Linear(n_embd, n_experts, bias=False, **kw)
        self.experts = nn.ModuleList(
            _Expert(n_embd, n_inner, tp_group=tp_group, **kw)
            for _ in range(self.n_local)
        )
        self.aux_loss: Optional[torch.Tensor] = None
        # capacity_factor > 0: Switch/GShard-style token dropping — each
        # expert processes at most ceil(cf * n_tok * k / n_experts) of
        # THIS RANK's assignments (position-priority within the stable
        # expert sort); dropped assignments contribute ZERO (the block's
        # residual stream carries the token through).  0 = exact
        # variable-size routing, nothing dropped.
        self.capacity_factor = float(capacity_factor)

    # ------------------------------------------------------------------
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B = x.shape[:-1]
        H = x.shape[-1]
        flat = x.reshape(-1, H)
        n_tok = flat.shape[0]

        logits = self.router(flat.float())
        probs = torch.softmax(logits, dim=-1)
        gates, top_idx = probs.topk(self.top_k, dim=-1)  # [n_tok, k]
        gates = gates / gates.sum(dim=-1, keepdim=True)

        # switch-style load-balancing aux loss: E * sum_e f_e * P_e
        with torch.no_grad():
            counts = torch.bincount(
                top_idx.reshape(-1), minlength=self.n_experts
            ).float()
            frac = counts / counts.sum().clamp(min=1)
        self.aux_loss = self.n_experts * (frac * probs.mean(dim=0)).sum()

        # dispatch: one row per (token, k) assignment, sorted by expert
        flat_idx = top_idx.reshape(-1)  # [n_tok*k]
        order = torch.argsort(flat_idx, stable=True)
        sorted_exp = flat_idx[order]
        per_expert = torch.bincount(sorted_exp, minlength=self.n_experts)
        n_assign = n_tok * self.top_k
        dropped = None
        if self.capacity_factor > 0:
            cap = int(-(-self.capacity_factor * n_assign // self.n_experts))
            if bool((per_expert > cap).any()):
                start = per_expert.cumsum(0) - per_expert  # exclusive cumsum
                within = torch.arange(n_assign, device=flat.device) - start[sorted_exp]
                keep = within < cap
                dropped = order[~keep]  # original assignment slots dropped
                order = order[keep]
                sorted_exp = sorted_exp[keep]
                per_expert = per_expert.clamp(max=cap)
        src_token = torch.div(order, self.top_k, rounding_mode="floor")
        routed = flat[src_token]  # [kept, H], grouped by expert

        # exchange: expert e lives on rank e // n_local
        out_splits = [
            int(per_expert[r * self.n_local : (r + 1) * self.n_local].sum())
            for r in range(self.ep_size)
        ]
        in_splits = self._exchange_counts(out_splits)
        arrived = all_to_all_var(routed, out_splits, in_splits, self.ep_group)

        # rows for a local expert arrive interleaved by SOURCE rank
        # (each source block is expert-sorted) — regroup to expert-major
        local_counts = self._exchange_expert_counts(per_expert)
        arrived_grouped, regroup_idx = self._regroup(arrived, local_counts)
        res = []
        off = 0
        for le in range(self.n_local):
            seg = int(local_counts[:, le].sum())
            res.append(self.experts[le](arrived_grouped[off : off + seg]))
            off += seg
        processed = torch.cat(res, dim=0) if res else arrived_grouped
        # undo regrouping, send results home, undo the expert sort — all
        # via inverse-permutation gathers (clean autograd)
        back = processed[torch.argsort(regroup_idx)]
        returned = all_to_all_var(back, in_splits, out_splits, self.ep_group)
        if dropped is None:
            unsorted = returned[torch.argsort(order)]
        else:
            # dropped assignment slots contribute zero
            unsorted = returned.new_zeros(n_assign, H).index_copy(
                0, order, returned
            )
        weighted = unsorted.view(n_tok, self.top_k, H) * gates.unsqueeze(-1).to(
            unsorted.dtype
        )
        return weighted.sum(dim=1).view(*B, H)

    # ------------------------------------------------------------------
    def _exchange_counts(self, out_splits: List[int]) -> List[int]:
        if self.ep_size == 1:
            return out_splits
        t = torch.tensor(out_splits, dtype=torch.long)
        gathered = [torch.zeros_like(t) for _ in range(self.ep_size)]
        dist.all_gather(gathered, t, group=self.ep_group)
        return [int(g[self.ep_rank]) for g in gathered]

    def _exchange_expert_counts(self, per_expert: torch.Tensor) -> torch.Tensor:
        """[ep_size, n_local]: how many rows each source rank sent to each
        of MY local experts."""
        if self.ep_size == 1:
            return per_expert.view(1, self.n_experts)[
                :, self.ep_rank * self.n_local : (self.ep_rank + 1) * self.n_local
            ]
        gathered = [torch.zeros_like(per_expert) for _ in range(self.ep_size)]
        dist.all_gather(gathered, per_expert, group=self.ep_group)
        lo = self.ep_rank * self.n_local
        return torch.stack([g[lo : lo + self.n_local] for g in gathered])

    @staticmethod
    def _regroup(arrived: torch.Tensor, local_counts: torch.Tensor):
        """arrived rows are source-major (then expert-minor within each
        source block); build the permutation to expert-major order."""
        lc = local_counts.cpu()
        ep, nl = lc.shape
        src_off = torch.zeros(ep, dtype=torch.long)
        src_off[1:] = lc.sum(dim=1).cumsum(0)[:-1]
        pieces = []
        for le in range(nl):
            for r in range(ep):
                start = int(src_off[r] + lc[r, :le].sum())
                pieces.append(torch.arange(start, start + int(lc[r, le])))
        idx = (torch.cat(pieces) if pieces else torch.arange(0)).to(arrived.device)
        return arrived[idx], idx
