"""Context parallelism (CP): the sequence axis sharded across GPUs for
attention.  Beyond reference parity (the reference has no CP).

MI355X-first design choice: with 288 GB of HBM3E per GPU, the all-gather-KV
formulation is the right first CP flavor — each rank holds its Q/K/V
sequence shard, all-gathers K and V over the CP group (one RCCL all-gather
each, backward = reduce-scatter of the KV grads via the autograd-aware
``All_Gather`` op), and runs the fused flash kernel on its Q shard against
the full K/V with the causal diagonal shifted by ``q_offset = cp_rank *
T_local`` (kernel support: csrc/attn.hip).  Activation memory for
everything OUTSIDE attention scales 1/cp; attention compute scales ~1/cp
on average (the causal upper ranks do more — ring/zigzag balancing is a
possible future refinement).
"""

from __future__ import annotations

import torch
import torch.distributed as dist

from ..core.comm import All_Gather
from ..ops.attention import attention

__all__ = [
    "context_parallel_attention",
    "ring_attention",
    "zigzag_ring_attention",
    "scatter_to_context",
    "zigzag_to_context",
    "zigzag_positions",
    "zigzag_clm_targets",
    "scatter_clm_targets",
    "cp_causal_lm_loss",
]


def _ws(group) -> int:
    if group is None or not dist.is_initialized():
        return 1
    return dist.get_world_size(group=group)


def scatter_to_context(x: torch.Tensor, cp_group, dim: int = 1) -> torch.Tensor:
    """Keep this rank's sequence shard (data is assumed replicated)."""
    world = _ws(cp_group)
    if world == 1:
        return x
    if x.shape[dim] % world != 0:
        raise ValueError(
            f"context parallel: dim {dim} ({x.shape[dim]}) must divide by cp={world}"
        )
    rank = dist.get_rank(group=cp_group)
    return x.chunk(world, dim=dim)[rank].contiguous()


def context_parallel_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cp_group,
    causal: bool = True,
) -> torch.Tensor:
    """Attention over the full sequence from per-rank [B, H, T/cp, D] shards.

    Returns this rank's [B, H, T/cp, D] output shard.  Differentiable:
    dK/dV flow back through the all-gather's reduce-scatter, dQ stays
    local.
    """
    world = _ws(cp_group)
    if world == 1:
        return attention(q, k, v, causal=causal)
    rank = dist.get_rank(group=cp_group)
    t_local = q.shape[-2]
    k_full = All_Gather.apply(k, cp_group, -2, "reduce_scatter")
    v_full = All_Gather.apply(v, cp_group, -2, "reduce_scatter")
    return attention(q, k_full, v_full, causal=causal, q_offset=rank * t_local)


# ---------------------------------------------------------------------------
# Ring attention: the memory-scalable CP flavor — K/V chunks rotate around
# the CP ring instead of being materialized in full on every rank.  One
# custom autograd Function owns BOTH ring schedules (forward: cp-1 hops of
# [K,V]; backward: cp hops of [K,V,dK,dV] with the grad accumulators
# traveling alongside their chunk until they return to the owner), so the
# P2P order is deterministic — no reliance on autograd's reverse-order
# engine to sequence collectives.  Per-chunk math is the composed fp32
# path (streaming log-sum-exp merge); the fused flash kernel can be slotted
# per chunk later.
# ---------------------------------------------------------------------------
class _SwapHandle:
    """In-flight ring exchange: keeps the SEND buffers referenced until
    wait() (``.contiguous()`` may have created temporaries; some
    backends do not pin send storage for async P2P)."""

    __slots__ = ("reqs", "recvs", "_sends")

    def __init__(self, reqs, recvs, sends):
        self.reqs, self.recvs, self._sends = reqs, recvs, sends

    def wait(self):
        for r in self.reqs:
            r.wait()
        return self.recvs


def _ring_swap_begin(tensors, rank, world, group):
    """Issue send-to-rank+1 / recv-from-rank-1 without waiting (overlap
    with compute: sends only READ the live buffers)."""
    nxt = (rank + 1) % world
    prv = (rank - 1 + world) % world
    sends = [t.contiguous() for t in tensors]
    ops = []
    recvs = []
    for t in sends:
        ops.append(dist.P2POp(dist.isend, t, peer=nxt, group=group))
    for t in tensors:
        buf = torch.empty(t.shape, dtype=t.dtype, device=t.device)
        recvs.append(buf)
        ops.append(dist.P2POp(dist.irecv, buf, peer=prv, group=group))
    return _SwapHandle(dist.batch_isend_irecv(ops), recvs, sends)


def _ring_swap(tensors, rank, world, group):
    """Send tensors to rank+1, receive the same shapes from rank-1."""
    if world == 1:
        return tensors
    return _ring_swap_begin(tensors, rank, world, group).wait()


def _partial(q32, k32, scale, causal_mode, Tl):
    """scores [B,H,Tl,Tl] for one chunk; causal_mode: 0 full, 1 diagonal."""
    s = torch.matmul(q32, k32.transpose(-2, -1)) * scale
    if causal_mode == 1:
        mask = torch.ones(Tl, Tl, dtype=torch.bool, device=s.device).tril()
        s = s.masked_fill(~mask, float("-inf"))
    return s


# ---------------------------------------------------------------------------
# Fused-flash per-chunk path (GPU, D=64, Tl%128==0): each ring hop runs the
# flash kernel on (Q shard, K/V chunk) and the chunk outputs merge with a
# streaming base-2 log-sum-exp — O(Tl·D) state, no T² score matrix.  The
# backward reuses the SAME kernels per chunk with the saved GLOBAL lse2
# (they recompute P = exp2(s·scale·log2e − lse2), which is exactly the
# globally-normalized probability of that chunk's keys).
# ---------------------------------------------------------------------------
def _ring_flash_ok(q: torch.Tensor, k: torch.Tensor) -> bool:
    from ..ops.attention import _flash_ok

    return (
        _flash_ok(q)
        and k.shape[-2] == q.shape[-2]
        and q.dtype == torch.bfloat16
    )


def _ring_flash_fwd_hop(q, kc, vc, scale, diag, out_run, lse_run):
    """One chunk through the fused kernel + streaming LSE merge.

    out_run [B,H,Tl,D] fp32 (normalized so far), lse_run [B,H,Tl,1] fp32
    base-2; returns the updated pair."""
    from .. import _C

    o_j = torch.empty_like(q)
    lse_j = _C.attn_fwd(q.contiguous(), kc, vc, o_j, scale, diag, 0)
    B, H, Tl, _ = q.shape
    lse_j = lse_j.view(B, H, Tl, 1)
    m = torch.maximum(lse_run, lse_j)
    a = torch.exp2(lse_run - m)
    b = torch.exp2(lse_j - m)
    denom = a + b
    out_run = (out_run * a + o_j.float() * b) / denom
    lse_run = m + torch.log2(denom)
    return out_run, lse_run


def _ring_flash_bwd_hop(q, kc, vc, out, dout, lse2_flat, scale, diag,
                        dqp, dkp, dvp):
    """Per-chunk backward via the fused kernels (writes dqp/dkp/dvp)."""
    from .. import _C

    _C.attn_bwd(q, kc, vc, out, dout, lse2_flat, dqp, dkp, dvp, scale,
                bool(diag), 0)


class _RingAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, group, causal):
        import math

        world = _ws(group)
        rank = dist.get_rank(group=group) if world > 1 else 0
        B, H, Tl, D = q.shape
        scale = 1.0 / math.sqrt(D)
        flash = _ring_flash_ok(q, k)
        kc, vc = k.contiguous(), v.contiguous()  # ring buffers must be
        if flash:
            qc = q.contiguous()
            out_run = torch.zeros(B, H, Tl, D, device=q.device)
            lse_run = torch.full((B, H, Tl, 1), float("-inf"), device=q.device)
            for s_hop in range(world):
                j = (rank - s_hop) % world
                hand = None
                if s_hop < world - 1:  # overlap the hop exchange w/ compute
                    hand = _ring_swap_begin([kc, vc], rank, world, group)
                if not causal or j <= rank:
                    out_run, lse_run = _ring_flash_fwd_hop(
                        qc, kc, vc, scale, 1 if (causal and j == rank) else 0,
                        out_run, lse_run,
                    )
                if hand is not None:
                    kc, vc = hand.wait()
            out = out_run.to(q.dtype)
            lse = lse_run.reshape(B * H, Tl)  # base-2, the kernels' format
        else:
            q32 = q.float()
            m = torch.full((B, H, Tl, 1), float("-inf"), device=q.device)
            l = torch.zeros(B, H, Tl, 1, device=q.device)
            acc = torch.zeros(B, H, Tl, D, device=q.device)
            for s_hop in range(world):
                j = (rank - s_hop) % world
                hand = None
                if s_hop < world - 1:  # overlap the hop exchange w/ compute
                    hand = _ring_swap_begin([kc, vc], rank, world, group)
                if not causal or j <= rank:
                    sc = _partial(q32, kc.float(), scale, 1 if (causal and j == rank) else 0, Tl)
                    # every processed chunk has >=1 unmasked key per row (the
                    # diagonal chunk includes key<=query within-chunk), so mj is
                    # finite and m_new is finite from the first hop; the initial
                    # m=-inf gives alpha=exp(-inf)=0 naturally.
                    mj = sc.amax(dim=-1, keepdim=True)
                    m_new = torch.maximum(m, mj)
                    alpha = torch.exp(m - m_new)
                    p = torch.exp(sc - m_new)
                    acc = acc * alpha + torch.matmul(p, vc.float())
                    l = l * alpha + p.sum(dim=-1, keepdim=True)
                    m = m_new
                if hand is not None:
                    kc, vc = hand.wait()
            out = (acc / l.clamp(min=1e-30)).to(q.dtype)
            lse = (m + torch.log(l.clamp(min=1e-30))).squeeze(-1)  # natural log
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.group = group
        ctx.causal = causal
        ctx.world = world
        ctx.rank = rank
        ctx.scale = scale
        ctx.flash = flash
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        group, causal = ctx.group, ctx.causal
        world, rank, scale = ctx.world, ctx.rank, ctx.scale
        B, H, Tl, D = q.shape
        kc, vc = k.contiguous(), v.contiguous()
        dk_acc = torch.zeros(B, H, Tl, D, device=q.device)
        dv_acc = torch.zeros(B, H, Tl, D, device=q.device)
        if ctx.flash:
            qc = q.contiguous()
            doutc = dout.contiguous().to(q.dtype)
            outc = out.contiguous()
            lse_flat = lse.contiguous()  # [B*H, Tl] base-2
            dq32 = torch.zeros(B, H, Tl, D, device=q.device)
            dqp = torch.empty_like(qc)
            dkp = torch.empty_like(qc)
            dvp = torch.empty_like(qc)
            for s_hop in range(world):
                j = (rank - s_hop) % world
                # kv only gets READ by the block compute: start its hop
                # exchange now and overlap; the grad accumulators are
                # WRITTEN during compute, so they travel afterwards
                # (split swap; P2P pair order kv-then-grads everywhere)
                hand_kv = None
                if world > 1:
                    hand_kv = _ring_swap_begin([kc, vc], rank, world, group)
                if not causal or j <= rank:
                    _ring_flash_bwd_hop(
                        qc, kc, vc, outc, doutc, lse_flat, scale,
                        1 if (causal and j == rank) else 0, dqp, dkp, dvp,
                    )
                    dq32 += dqp.float()
                    dk_acc += dkp.float()
                    dv_acc += dvp.float()
                if hand_kv is not None:
                    hand_g = _ring_swap_begin(
                        [dk_acc, dv_acc], rank, world, group)
                    kc, vc = hand_kv.wait()
                    dk_acc, dv_acc = hand_g.wait()
            return (
                dq32.to(q.dtype),
                dk_acc.to(k.dtype),
                dv_acc.to(v.dtype),
                None,
                None,
            )
        q32, dout32, out32 = q.float(), dout.float(), out.float()
        delta = (dout32 * out32).sum(dim=-1, keepdim=True)  # [B,H,Tl,1]
        lse_ = lse.unsqueeze(-1)
        dq = torch.zeros_like(q32)
        # grad accumulators travel with their chunk for a full cycle
        for s_hop in range(world):
            j = (rank - s_hop) % world
            if not causal or j <= rank:
                sc = _partial(q32, kc.float(), scale, 1 if (causal and j == rank) else 0, Tl)
                p = torch.exp(sc - lse_)  # normalized probs of this chunk
                dp = torch.matmul(dout32, vc.float().transpose(-2, -1))
                ds = p * (dp - delta) * scale
                dq = dq + torch.matmul(ds, kc.float())
                dk_acc = dk_acc + torch.matmul(ds.transpose(-2, -1), q32)
                dv_acc = dv_acc + torch.matmul(p.transpose(-2, -1), dout32)
            # rotate kv + their grads one more hop; after `world` hops the
            # accumulators are back at the chunk owner (kv pre-rotated
            # before compute in the flash branch above; here the hop is
            # synchronous — the fp32 branch is the small-shape fallback)
            kc, vc, dk_acc, dv_acc = _ring_swap(
                [kc, vc, dk_acc, dv_acc], rank, world, group
            )
        return (
            dq.to(q.dtype),
            dk_acc.to(k.dtype),
            dv_acc.to(v.dtype),
            None,
            None,
        )


# ---------------------------------------------------------------------------
# Zigzag (load-balanced) ring attention: the sequence is split into 2·cp
# chunks and rank r holds chunks (r, 2cp−1−r) — under the causal mask every
# rank then processes the SAME number of (q-half, kv-half) blocks per ring
# cycle (plain ring gives rank r work ∝ r+1).  Per block the fused flash
# kernel runs at half-shard size (fp32 composed fallback on CPU); merges
# are streaming per q-half.
# ---------------------------------------------------------------------------
def _zz_ids(rank: int, world: int):
    return (rank, 2 * world - 1 - rank)


def zigzag_to_context(x: torch.Tensor, cp_group, dim: int = 1) -> torch.Tensor:
    """Keep this rank's ZIGZAG sequence shard: chunks (r, 2cp−1−r) of the
    2·cp-way split, concatenated."""
    world = _ws(cp_group)
    if world == 1:
        return x
    if x.shape[dim] % (2 * world) != 0:
        raise ValueError(
            f"zigzag CP: dim {dim} ({x.shape[dim]}) must divide by 2*cp={2*world}"
        )
    rank = dist.get_rank(group=cp_group)
    lo, hi = _zz_ids(rank, world)
    ch = x.chunk(2 * world, dim=dim)
    return torch.cat([ch[lo], ch[hi]], dim=dim).contiguous()


def zigzag_positions(T: int, cp_group, device) -> torch.Tensor:
    """Global position ids of this rank's zigzag shard (for wpe)."""
    world = _ws(cp_group)
    rank = dist.get_rank(group=cp_group) if world > 1 else 0
    h = T // (2 * world)
    lo, hi = _zz_ids(rank, world)
    return torch.cat([
        torch.arange(lo * h, (lo + 1) * h, device=device),
        torch.arange(hi * h, (hi + 1) * h, device=device),
    ])


def zigzag_clm_targets(labels: torch.Tensor, cp_group, ignore_index: int = -100):
    """Shift the FULL label sequence, then keep the zigzag shard."""
    from ..ops import shift_labels

    return zigzag_to_context(shift_labels(labels, ignore_index), cp_group, dim=1)


class _ZigzagRingAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, group, causal):
        import math

        world = _ws(group)
        rank = dist.get_rank(group=group) if world > 1 else 0
        B, H, Tl, D = q.shape
        assert Tl % 2 == 0, "zigzag shard must hold two equal half-chunks"
        h = Tl // 2
        scale = 1.0 / math.sqrt(D)
        flash = _ring_flash_ok(q, k) and (h % 128 == 0)
        cq = _zz_ids(rank, world)
        kc, vc = k.contiguous(), v.contiguous()
        if flash:
            qc = q.contiguous()
            qh = [qc[:, :, :h].contiguous(), qc[:, :, h:].contiguous()]
            out_run = [torch.zeros(B, H, h, D, device=q.device) for _ in range(2)]
            lse_run = [
                torch.full((B, H, h, 1), float("-inf"), device=q.device)
                for _ in range(2)
            ]
        else:
            q32 = q.float()
            m = torch.full((B, H, Tl, 1), float("-inf"), device=q.device)
            l = torch.zeros(B, H, Tl, 1, device=q.device)
            acc = torch.zeros(B, H, Tl, D, device=q.device)
        for s_hop in range(world):
            j = (rank - s_hop) % world
            ck = _zz_ids(j, world)
            hand_kv = None
            if s_hop < world - 1:  # overlap the hop exchange w/ compute
                hand_kv = _ring_swap_begin([kc, vc], rank, world, group)
            for qi in range(2):
                for ki in range(2):
                    if causal and ck[ki] > cq[qi]:
                        continue
                    diag = 1 if (causal and ck[ki] == cq[qi]) else 0
                    ksl = kc[:, :, ki * h : (ki + 1) * h]
                    vsl = vc[:, :, ki * h : (ki + 1) * h]
                    if flash:
                        out_run[qi], lse_run[qi] = _ring_flash_fwd_hop(
                            qh[qi], ksl.contiguous(), vsl.contiguous(), scale,
                            diag, out_run[qi], lse_run[qi],
                        )
                    else:
                        sl = slice(qi * h, (qi + 1) * h)
                        sc = _partial(q32[:, :, sl], ksl.float(), scale, diag, h)
                        mj = sc.amax(dim=-1, keepdim=True)
                        m_new = torch.maximum(m[:, :, sl], mj)
                        alpha = torch.exp(m[:, :, sl] - m_new)
                        p = torch.exp(sc - m_new)
                        acc[:, :, sl] = acc[:, :, sl] * alpha + torch.matmul(
                            p, vsl.float())
                        l[:, :, sl] = l[:, :, sl] * alpha + p.sum(
                            dim=-1, keepdim=True)
                        m[:, :, sl] = m_new
            if hand_kv is not None:
                kc, vc = hand_kv.wait()
        if flash:
            out = torch.cat(out_run, dim=2).to(q.dtype)
            lse = torch.cat(lse_run, dim=2).reshape(B * H, Tl)  # base-2
        else:
            out = (acc / l.clamp(min=1e-30)).to(q.dtype)
            lse = (m + torch.log(l.clamp(min=1e-30))).squeeze(-1)  # natural
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.group, ctx.causal = group, causal
        ctx.world, ctx.rank, ctx.scale = world, rank, scale
        ctx.flash = flash
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        group, causal = ctx.group, ctx.causal
        world, rank, scale = ctx.world, ctx.rank, ctx.scale
        B, H, Tl, D = q.shape
        h = Tl // 2
        cq = _zz_ids(rank, world)
        kc, vc = k.contiguous(), v.contiguous()
        dk_acc = torch.zeros(B, H, Tl, D, device=q.device)
        dv_acc = torch.zeros(B, H, Tl, D, device=q.device)
        if ctx.flash:
            qc = q.contiguous()
            doutc = dout.contiguous().to(q.dtype)
            outc = out.contiguous()
            lse4 = lse.view(B, H, Tl)
            dq32 = torch.zeros(B, H, Tl, D, device=q.device)
            dqp = torch.empty(B, H, h, D, device=q.device, dtype=q.dtype)
            dkp = torch.empty_like(dqp)
            dvp = torch.empty_like(dqp)
            for s_hop in range(world):
                j = (rank - s_hop) % world
                ck = _zz_ids(j, world)
                if world > 1:  # pre-rotate kv under the block compute
                    hand_kv = _ring_swap_begin([kc, vc], rank, world, group)
                for qi in range(2):
                    for ki in range(2):
                        if causal and ck[ki] > cq[qi]:
                            continue
                        diag = 1 if (causal and ck[ki] == cq[qi]) else 0
                        qsl = slice(qi * h, (qi + 1) * h)
                        ksl = slice(ki * h, (ki + 1) * h)
                        _ring_flash_bwd_hop(
                            qc[:, :, qsl].contiguous(),
                            kc[:, :, ksl].contiguous(),
                            vc[:, :, ksl].contiguous(),
                            outc[:, :, qsl].contiguous(),
                            doutc[:, :, qsl].contiguous(),
                            lse4[:, :, qsl].reshape(B * H, h).contiguous(),
                            scale, diag, dqp, dkp, dvp,
                        )
                        dq32[:, :, qsl] += dqp.float()
                        dk_acc[:, :, ksl] += dkp.float()
                        dv_acc[:, :, ksl] += dvp.float()
                if world > 1:
                    hand_g = _ring_swap_begin(
                        [dk_acc, dv_acc], rank, world, group)
                    kc, vc = hand_kv.wait()
                    dk_acc, dv_acc = hand_g.wait()
            return (dq32.to(q.dtype), dk_acc.to(k.dtype), dv_acc.to(v.dtype),
                    None, None)
        q32, dout32, out32 = q.float(), dout.float(), out.float()
        delta = (dout32 * out32).sum(dim=-1, keepdim=True)
        lse_ = lse.unsqueeze(-1)
        dq = torch.zeros_like(q32)
        for s_hop in range(world):
            j = (rank - s_hop) % world
            ck = _zz_ids(j, world)
            for qi in range(2):
                for ki in range(2):
                    if causal and ck[ki] > cq[qi]:
                        continue
                    diag = 1 if (causal and ck[ki] == cq[qi]) else 0
                    qsl = slice(qi * h, (qi + 1) * h)
                    ksl = slice(ki * h, (ki + 1) * h)
                    sc = _partial(q32[:, :, qsl], kc[:, :, ksl].float(), scale,
                                  diag, h)
                    p = torch.exp(sc - lse_[:, :, qsl])
                    dp = torch.matmul(dout32[:, :, qsl],
                                      vc[:, :, ksl].float().transpose(-2, -1))
                    ds = p * (dp - delta[:, :, qsl]) * scale
                    dq[:, :, qsl] += torch.matmul(ds, kc[:, :, ksl].float())
                    dk_acc[:, :, ksl] += torch.matmul(
                        ds.transpose(-2, -1), q32[:, :, qsl])
                    dv_acc[:, :, ksl] += torch.matmul(
                        p.transpose(-2, -1), dout32[:, :, qsl])
            kc, vc, dk_acc, dv_acc = _ring_swap(
                [kc, vc, dk_acc, dv_acc], rank, world, group)
        return (dq.to(q.dtype), dk_acc.to(k.dtype), dv_acc.to(v.dtype),
                None, None)


def zigzag_ring_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cp_group,
    causal: bool = True,
) -> torch.Tensor:
    """Load-balanced ring CP attention over ZIGZAG shards (chunks
    (r, 2cp−1−r) of a 2·cp split — use ``zigzag_to_context`` /
    ``zigzag_positions`` for the inputs).  Every rank processes the same
    block count under the causal mask; per-block compute is the fused
    flash kernel on GPU."""
    return _ZigzagRingAttention.apply(q, k, v, cp_group, causal)


def ring_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cp_group,
    causal: bool = True,
) -> torch.Tensor:
    """Ring-form context-parallel attention over [B, H, T/cp, D] shards.

    Peak memory is O(T/cp) per rank for K/V (vs the full-KV all-gather of
    ``context_parallel_attention``) at the cost of cp-1 (forward) / cp
    (backward) ring exchanges.  Differentiable; the comm schedule is
    owned by the op, so pass a DEDICATED cp communicator."""
    if _ws(cp_group) == 1:
        return attention(q, k, v, causal=causal)
    return _RingAttention.apply(q, k, v, cp_group, causal)


# ---------------------------------------------------------------------------
# CP training helpers: with every activation sequence-sharded end to end
# (no gather anywhere), every parameter gradient is a PARTITION of the full
# gradient — so a DataParallel wrap over the cp group with the loss
# pre-scaled by cp gives exactly ∇L after its MEAN reduction:
#   grad = (1/cp) Σ_r ∇(cp · L_r) = Σ_r ∇L_r = ∇L.
# The local loss L_r is CE summed over this rank's shard divided by the
# GLOBAL valid-token count (all-reduced, non-differentiable).
# ---------------------------------------------------------------------------
def scatter_clm_targets(labels: torch.Tensor, cp_group, ignore_index: int = -100):
    """Shift the FULL label sequence for causal LM (so cross-shard next
    tokens are correct), then keep this rank's shard."""
    from ..ops import shift_labels

    return scatter_to_context(shift_labels(labels, ignore_index), cp_group, dim=1)


def cp_causal_lm_loss(
    logits_shard: torch.Tensor,
    targets_shard: torch.Tensor,
    cp_group,
    ignore_index: int = -100,
):
    """Returns (loss_for_backward, true_loss_detached).

    ``loss_for_backward`` is cp * CE_sum(shard) / N_total — backward it on
    every rank and let a DataParallel(cp_group) MEAN-reduce the gradients.
    ``true_loss`` is the exact global mean CE for logging.
    """
    world = _ws(cp_group)
    ce_sum = torch.nn.functional.cross_entropy(
        logits_shard.float().reshape(-1, logits_shard.shape[-1]),
        targets_shard.reshape(-1),
        ignore_index=ignore_index,
        reduction="sum",
    )
    n_local = (targets_shard != ignore_index).sum()
    n_total = n_local.clone()
    if world > 1:
        dist.all_reduce(n_total, group=cp_group)
    loss_bwd = ce_sum / n_total.clamp(min=1) * world
    with torch.no_grad():
        true = ce_sum.detach().clone()
        if world > 1:
            dist.all_reduce(true, group=cp_group)
        true = true / n_total.clamp(min=1)
    return loss_bwd, true
