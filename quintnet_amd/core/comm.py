"""Autograd-aware collectives and pipeline P2P over RCCL/xGMI.

Functional parity with reference core/communication.py (Send/Recv at
:46-204, pipeline_communicate at :207-296, bidirectional at :299-371,
All_Gather :374-475, All_Reduce :478-535, ReduceScatter :538-600), with
MI355X-native changes:

* Pipeline P2P uses statically-known tensor shapes (the schedules always
  know them) — no per-message ndims/shape negotiation round-trips.
* No ``torch.cuda.synchronize()`` after P2P batches.  ``Work.wait()`` on
  the returned reqs orders the RCCL stream against the compute stream;
  the device never has to drain.
* Each PP pair is a single dedicated xGMI link (~153 GB/s) so grouped
  batched isend/irecv is all that is needed; overlap comes from issuing
  the batch before dependent compute.
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

__all__ = [
    "Send",
    "Recv",
    "All_Gather",
    "All_Reduce",
    "ReduceScatter",
    "pipeline_communicate",
    "bidirectional_pipeline_communicate",
    "ring_send",
    "ring_recv",
    "send_tensor",
    "recv_tensor",
]



def _ws(group) -> int:
    """group world size; 1 when torch.distributed is uninitialized."""
    if not dist.is_initialized():
        return 1
    return dist.get_world_size(group=group)

# ---------------------------------------------------------------------------
# Autograd point-to-point (shape negotiated once via metadata — used by the
# generic API; the PP schedules use the static-shape fast path below).
# ---------------------------------------------------------------------------
class Send(torch.autograd.Function):
    """Send ``x`` to ``dst`` in forward; receive its grad in backward."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, dst: int, group=None):
        ctx.dst = dst
        ctx.group = group
        ctx.shape = x.shape
        ctx.dtype = x.dtype
        ctx.device = x.device
        dist.send(x.contiguous(), dst=dst, group=group)
        return torch.zeros(1, device=x.device, dtype=x.dtype)

    @staticmethod
    def backward(ctx, grad_output):
        grad = torch.empty(ctx.shape, dtype=ctx.dtype, device=ctx.device)
        dist.recv(grad, src=ctx.dst, group=ctx.group)
        return grad, None, None


class Recv(torch.autograd.Function):
    """Receive a tensor from ``src`` in forward; send grad back in backward."""

    @staticmethod
    def forward(ctx, buffer: torch.Tensor, src: int, group=None):
        ctx.src = src
        ctx.group = group
        dist.recv(buffer, src=src, group=group)
        return buffer

    @staticmethod
    def backward(ctx, grad_output):
        dist.send(grad_output.contiguous(), dst=ctx.src, group=ctx.group)
        return None, None, None


def send_tensor(x: torch.Tensor, dst: int, group=None) -> None:
    dist.send(x.contiguous(), dst=dst, group=group)


def recv_tensor(
    shape: Tuple[int, ...], dtype: torch.dtype, device: torch.device, src: int, group=None
) -> torch.Tensor:
    buf = torch.empty(shape, dtype=dtype, device=device)
    dist.recv(buf, src=src, group=group)
    return buf


# ---------------------------------------------------------------------------
# Autograd collectives for TP
# ---------------------------------------------------------------------------
class All_Gather(torch.autograd.Function):
    """Gather shards along ``dim`` across ``group``.

    forward: all_gather + cat(dim).  backward: take my slice (default) or
    reduce_scatter when ``backward_mode='reduce_scatter'`` (the SP seam —
    reference core/communication.py:374-475).
    """

    @staticmethod
    def forward(ctx, x: torch.Tensor, group=None, dim: int = -1, backward_mode: str = "slice"):
        world = _ws(group)
        ctx.group = group
        ctx.dim = dim
        ctx.world = world
        ctx.backward_mode = backward_mode
        ctx.rank = dist.get_rank(group=group) if world > 1 else 0
        if world == 1:
            return x
        x = x.contiguous()
        out_shape = list(x.shape)
        out_shape[dim] *= world
        # Gather into one flat buffer then view — single RCCL all_gather.
        gathered = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(gathered, x, group=group)
        return torch.cat(gathered, dim=dim)

    @staticmethod
    def backward(ctx, grad_output):
        if ctx.world == 1:
            return grad_output, None, None, None
        if ctx.backward_mode == "reduce_scatter":
            chunks = list(grad_output.contiguous().chunk(ctx.world, dim=ctx.dim))
            chunks = [c.contiguous() for c in chunks]
            out = torch.empty_like(chunks[ctx.rank])
            dist.reduce_scatter(out, chunks, op=dist.ReduceOp.SUM, group=ctx.group)
            return out, None, None, None
        # default: each rank keeps the grad slice of its own shard
        return (
            grad_output.chunk(ctx.world, dim=ctx.dim)[ctx.rank].contiguous(),
            None,
            None,
            None,
        )


class All_Reduce(torch.autograd.Function):
    """Sum across group in forward; identity backward (RowParallel output).

    Writes a fresh buffer (reference quirk §8.5: it aliased its input).
    """

    @staticmethod
    def forward(ctx, x: torch.Tensor, group=None):
        ctx.group = group
        if _ws(group) == 1:
            return x
        out = x.clone() if x.is_contiguous() else x.contiguous()
        dist.all_reduce(out, op=dist.ReduceOp.SUM, group=group)
        return out

    @staticmethod
    def backward(ctx, grad_output):
        return grad_output, None


class _CopyToGroup(torch.autograd.Function):
    """Identity forward; all-reduce(SUM) of grad in backward.

    The ``f`` operator of Megatron TP: input fed to column-parallel
    layers whose grads must be summed over the TP group.
    """

    @staticmethod
    def forward(ctx, x: torch.Tensor, group=None):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad_output):
        if _ws(ctx.group) > 1:
            grad_output = grad_output.contiguous()
            dist.all_reduce(grad_output, op=dist.ReduceOp.SUM, group=ctx.group)
        return grad_output, None


def copy_to_group(x: torch.Tensor, group=None) -> torch.Tensor:
    return _CopyToGroup.apply(x, group)


class _ScatterToSequence(torch.autograd.Function):
    """fwd: keep this rank's sequence shard; bwd: all-gather the grads.

    The entry operator of Megatron sequence parallelism (splits the
    replicated embedding output across the TP group).
    """

    @staticmethod
    def forward(ctx, x: torch.Tensor, group=None, dim: int = 1):
        world = _ws(group)
        ctx.group = group
        ctx.dim = dim
        ctx.world = world
        if world == 1:
            return x
        rank = dist.get_rank(group=group)
        return x.chunk(world, dim=dim)[rank].contiguous()

    @staticmethod
    def backward(ctx, grad_output):
        if ctx.world == 1:
            return grad_output, None, None
        gathered = [torch.empty_like(grad_output) for _ in range(ctx.world)]
        dist.all_gather(gathered, grad_output.contiguous(), group=ctx.group)
        return torch.cat(gathered, dim=ctx.dim), None, None


def scatter_to_sequence(x: torch.Tensor, group=None, dim: int = 1) -> torch.Tensor:
    return _ScatterToSequence.apply(x, group, dim)


class ReduceScatter(torch.autograd.Function):
    """forward: reduce_scatter(SUM) along dim; backward: all_gather.

    Defined for sequence-parallel drop-in (reference K12, unused by the
    current layers).
    """

    @staticmethod
    def forward(ctx, x: torch.Tensor, group=None, dim: int = -1):
        world = _ws(group)
        ctx.group = group
        ctx.dim = dim
        ctx.world = world
        if world == 1:
            return x
        chunks = [c.contiguous() for c in x.chunk(world, dim=dim)]
        out = torch.empty_like(chunks[0])
        dist.reduce_scatter(out, chunks, op=dist.ReduceOp.SUM, group=group)
        return out

    @staticmethod
    def backward(ctx, grad_output):
        if ctx.world == 1:
            return grad_output, None, None
        gathered = [torch.empty_like(grad_output) for _ in range(ctx.world)]
        dist.all_gather(gathered, grad_output.contiguous(), group=ctx.group)
        return torch.cat(gathered, dim=ctx.dim), None, None


# ---------------------------------------------------------------------------
# Pipeline P2P — static shapes, batched isend/irecv, no device sync
# ---------------------------------------------------------------------------
def _neighbor(group_ranks: List[int], my_pos: int, delta: int) -> Optional[int]:
    pos = my_pos + delta
    if pos < 0 or pos >= len(group_ranks):
        return None
    return group_ranks[pos]


def pipeline_communicate(
    operation: str,
    pp_rank: int,
    pp_size: int,
    group_ranks: List[int],
    tensor: Optional[torch.Tensor] = None,
    shapes: Optional[Tuple[int, ...]] = None,
    dtype: Optional[torch.dtype] = None,
    device: Optional[torch.device] = None,
    group=None,
) -> Optional[torch.Tensor]:
    """One P2P op on the pp axis: recv_forward | send_forward |
    recv_backward | send_backward.

    Reference parity: core/communication.py:207-296, minus the trailing
    ``torch.cuda.synchronize()`` (replaced by ``req.wait()`` stream
    ordering) and minus shape negotiation (shapes are static).
    """
    if operation == "recv_forward":
        if pp_rank == 0:
            return None
        src = _neighbor(group_ranks, pp_rank, -1)
        buf = torch.empty(shapes, dtype=dtype, device=device, requires_grad=True)
        op = dist.P2POp(dist.irecv, buf, peer=src, group=group)
    elif operation == "send_forward":
        if pp_rank == pp_size - 1:
            return None
        dst = _neighbor(group_ranks, pp_rank, +1)
        op = dist.P2POp(dist.isend, tensor.contiguous(), peer=dst, group=group)
        buf = None
    elif operation == "recv_backward":
        if pp_rank == pp_size - 1:
            return None
        src = _neighbor(group_ranks, pp_rank, +1)
        buf = torch.empty(shapes, dtype=dtype, device=device)
        op = dist.P2POp(dist.irecv, buf, peer=src, group=group)
    elif operation == "send_backward":
        if pp_rank == 0:
            return None
        dst = _neighbor(group_ranks, pp_rank, -1)
        op = dist.P2POp(dist.isend, tensor.contiguous(), peer=dst, group=group)
        buf = None
    else:
        raise ValueError(f"unknown pipeline op {operation!r}")

    reqs = dist.batch_isend_irecv([op])
    for r in reqs:
        r.wait()
    return buf


def bidirectional_pipeline_communicate(
    operation: str,
    pp_rank: int,
    pp_size: int,
    group_ranks: List[int],
    send_tensor: torch.Tensor,
    recv_shapes: Tuple[int, ...],
    dtype: torch.dtype,
    device: torch.device,
    group=None,
) -> Optional[torch.Tensor]:
    """Paired isend+irecv for the 1F1B steady state.

    operation: 'send_fwd_recv_bwd' (send act to next, recv grad from
    next) or 'send_bwd_recv_fwd' (send grad to prev, recv act from prev).
    Reference parity: core/communication.py:299-371.
    """
    is_fwd = operation == "send_fwd_recv_bwd"
    if is_fwd and pp_rank == pp_size - 1:
        return None
    if not is_fwd and pp_rank == 0:
        return None
    peer = _neighbor(group_ranks, pp_rank, +1 if is_fwd else -1)
    recv_buf = torch.empty(
        recv_shapes, dtype=dtype, device=device, requires_grad=not is_fwd
    )
    ops = [
        dist.P2POp(dist.isend, send_tensor.contiguous(), peer=peer, group=group),
        dist.P2POp(dist.irecv, recv_buf, peer=peer, group=group),
    ]
    reqs = dist.batch_isend_irecv(ops)
    for r in reqs:
        r.wait()
    return recv_buf


# ---------------------------------------------------------------------------
# Ring P2P for the interleaved-1F1B schedule: peers wrap around (rank p-1
# sends chunk boundaries forward to rank 0).  Sends are NON-blocking (the
# caller keeps the tensor alive and waits the returned reqs at step end) —
# with sends posted before the next blocking recv, the schedule is
# deadlock-free: the topologically-earliest unsatisfied recv's producer has
# already posted its isend (program order: recv -> compute -> isend).
# ---------------------------------------------------------------------------
def ring_send(
    tensor: torch.Tensor,
    pp_rank: int,
    pp_size: int,
    group_ranks: List[int],
    delta: int,
    group=None,
):
    dst = group_ranks[(pp_rank + delta) % pp_size]
    op = dist.P2POp(dist.isend, tensor.contiguous(), peer=dst, group=group)
    return dist.batch_isend_irecv([op])


def ring_recv(
    pp_rank: int,
    pp_size: int,
    group_ranks: List[int],
    delta: int,
    shapes,
    dtype,
    device,
    group=None,
    requires_grad: bool = False,
) -> torch.Tensor:
    src = group_ranks[(pp_rank + delta) % pp_size]
    buf = torch.empty(shapes, dtype=dtype, device=device, requires_grad=requires_grad)
    reqs = dist.batch_isend_irecv([dist.P2POp(dist.irecv, buf, peer=src, group=group)])
    for r in reqs:
        r.wait()
    return buf
