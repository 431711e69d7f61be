"""Linear (GEMM) op with fused bias/activation epilogue.

Per-shape measured dispatch between the hand-written CDNA4 MFMA GEMMs
(csrc/gemm.hip — NT form C[M,N] = A[M,K]·B[N,K]^T matching nn.Linear's
[out,in] layout, fused bias/GELU/ReLU epilogues, incl. a persistent
cross-tile-pipelined variant) and TunableOp-tuned hipBLASLt.  The tuned
library currently wins the plain GPT-2 forward shapes (see the policy
table below); the custom split-K TN kernel serves the wgrad shapes it
wins; dgrad is library.  MI355X design split: hand-written kernels own
the FUSED hot ops (attention, LN, CE, AdamW, wgrad), the vendor library
serves plain GEMMs.

Replaces the implicit cuBLAS GEMMs at reference
parallelism/tensor_parallel/layers.py:119,211 and utils/GPT2/*.
"""

from __future__ import annotations

from typing import Optional

import torch

from . import _backend

__all__ = ["linear", "LinearFunction", "defer_wgrads", "flush_deferred_wgrads"]

_ACT_NONE = 0
_ACT_GELU = 1
_ACT_RELU = 2
_ACT_MAP = {None: _ACT_NONE, "none": _ACT_NONE, "gelu": _ACT_GELU, "relu": _ACT_RELU}


def _gelu_tanh(x: torch.Tensor) -> torch.Tensor:
    return torch.nn.functional.gelu(x, approximate="tanh")


import os

# Forward dispatch policy (A/B-measured, profiles/gemm_tuned.log r2):
# per-shape comparison of the hand-written MFMA kernels vs hipBLASLt at
# its TunableOp-tuned best (M=16384, bench micro 16):
#   c_attn (N2304 K768):  library 888 vs custom 676
#   c_proj (N768  K768):  library 645 vs custom 557
#   c_fc   (N3072 K768):  library 619 vs custom 546 (GELU-fused)
#   mlp    (N768  K3072): library 1131 vs custom 864
# The tuned library wins every bench shape, so "auto" keeps it — per the
# MI355X design split the hand-written kernels own the FUSED hot ops
# (attention, LN, CE, AdamW, wgrad) and the library serves plain GEMMs.
# QN_GEMM_FWD=custom forces the MFMA kernels (per-shape best variant);
# =library forces hipBLASLt everywhere.
_FWD_MODE = os.environ.get("QN_GEMM_FWD", "auto")
_WGRAD_MODE = os.environ.get("QN_WGRAD", "auto")
# GELU-epilogue GEMM (csrc/blaslt.cpp): one hipBLASLt kernel computes
# gemm+bias+gelu and stores the pre-activation aux for the fused
# act_bwd — replaces the separate eager GELU pass (~90 us/layer).
# EXPERIMENTAL, default off: this ROCm 7.2 hipBLASLt build returns NO
# algo for the GELU_AUX_BIAS bf16 epilogue on gfx950 (heuristic count
# 0 — profiles/gelu_epi_diag note), so the branch raises and falls back
# to the eager GELU; revisit when the library gains the epilogue.
_GELU_EPI = os.environ.get("QN_GELU_EPI", "0") == "1"
_gelu_epi_broken = False

# ---------------------------------------------------------------------------
# Deferred weight gradients (zero-bubble building block): with the mode
# on, LinearFunction.backward computes grad_x (the critical path that
# feeds the upstream pipeline stage) immediately but QUEUES the weight
# GEMM; flush_deferred_wgrads() runs the queued dW GEMMs later — e.g.
# while a pipeline stage would otherwise idle in a bubble or block on a
# P2P wait — and ACCUMULATES into param.grad exactly as autograd would
# have (sum over micro-batches; DDP bucket views receive the adds, so
# finalize_gradients() reduces complete gradients).  Requires hook-fired
# bucket reduction to be OFF during backward (the hooks key on autograd
# accumulation, which deferral bypasses); DataParallel.finalize_gradients
# launches every bucket itself, which is the ZB reduction path.
_DEFER_WGRADS = False
_DEFERRED: list = []


class _DeferScope:
    def __enter__(self):
        defer_wgrads(True)
        return self

    def __exit__(self, *a):
        defer_wgrads(False)
        return False


def defer_wgrads(on: bool = True):
    """Toggle deferred-dW mode.  Usable as a context manager:
    ``with defer_wgrads.scope(): loss.backward()`` then
    ``flush_deferred_wgrads()`` before any grad consumer."""
    global _DEFER_WGRADS
    _DEFER_WGRADS = bool(on)


defer_wgrads.scope = _DeferScope


def flush_deferred_wgrads() -> int:
    """Run every queued weight-gradient GEMM, accumulating into
    ``param.grad`` (creating it when absent).  Returns the number of
    flushed entries.  MUST run before grad reduction / clipping /
    optimizer step."""
    global _DEFERRED
    queue, _DEFERRED = _DEFERRED, []
    with torch.no_grad():  # the queued activations carry autograd history
        for g, x2d, weight in queue:
            dw = _wgrad(g, x2d)
            if weight.grad is None:
                weight.grad = dw
            else:
                weight.grad += dw
    return len(queue)


def _wgrad(g: torch.Tensor, x2d: torch.Tensor) -> torch.Tensor:
    """The same per-shape dW dispatch backward() uses."""
    if (
        _backend.use_native(g)
        and _backend.has_ext()
        and g.dtype == torch.bfloat16
        and g.shape[1] % 128 == 0
        and x2d.shape[1] % 128 == 0
        and g.shape[0] % 64 == 0
        and (
            (g.shape[1] // 128) * (x2d.shape[1] // 128) < 128
            or _WGRAD_MODE == "custom"
        )
        and _WGRAD_MODE != "library"
    ):
        return _backend.ext().wgrad_tn(g.contiguous(), x2d.contiguous())
    return g.t() @ x2d


def _custom_wins_shape(m: int, n: int, k: int) -> bool:
    # r2 final A/B (profiles/gemm_tuned.log): against a FULLY TunableOp-
    # tuned hipBLASLt, the library wins every GPT-2 bench shape (the
    # earlier narrow-N "win" was against a badly-tuned library run —
    # library c_proj varies 337-662 TF with tuning quality, which is
    # exactly why the tuned CSV is persisted under profiles/).  Auto
    # keeps the library; QN_GEMM_FWD=custom forces the MFMA kernels.
    return False


def _custom_mode(m: int, n: int, k: int) -> int:
    # per-shape best custom variant: persistent 8-phase for eligible big
    # shapes with wide N or deep K; 128² two-buffer kernel otherwise
    if m % 256 == 0 and n % 256 == 0 and k % 128 == 0 and (
        n >= 3072 or k >= 2048
    ):
        return 5
    return 1


def _native_ok(x: torch.Tensor, weight: torch.Tensor) -> bool:
    # The MFMA kernel covers bf16 with K a multiple of 8 (staging width);
    # anything else takes the library GEMM path.
    return (
        _backend.has_ext()
        and x.is_cuda
        and x.dtype == torch.bfloat16
        and weight.dtype == torch.bfloat16
        and weight.shape[1] % 8 == 0
    )


def _fp8_ok(x2d: torch.Tensor, weight: torch.Tensor) -> bool:
    return (
        x2d.is_cuda
        and x2d.dtype == torch.bfloat16
        and hasattr(torch, "float8_e4m3fn")
        and x2d.shape[0] % 16 == 0
        and weight.shape[0] % 16 == 0
        and weight.shape[1] % 32 == 0
    )


_FP8_MAX = 448.0
from torch.utils.weak import WeakTensorKeyDictionary

_FP8_STATE = WeakTensorKeyDictionary()  # weight -> cached fp8 + delayed scale


def _fp8_linear(x2d: torch.Tensor, weight: torch.Tensor, bias):
    """Forward-only OCP e4m3 GEMM via hipBLASLt (torch._scaled_mm) —
    measured 1287 TF on the GPT-2 c_fc shape vs 833 bf16 (probe:
    tools/probe_fp8.py).  Backward stays bf16.  Overheads amortized the
    standard way: the fp8 weight is cached until the parameter version
    changes (one cast per optimizer step), and activations use DELAYED
    scaling (cast with the previous step's amax; this step's amax is
    computed asynchronously and becomes the next scale) so the only
    per-call extra work is the single cast pass the fp8 GEMM requires."""
    st = _FP8_STATE.setdefault(weight, {})
    wver = getattr(weight, "_version", None)
    if st.get("wver") != wver or st.get("w8") is None:
        sb = (weight.detach().abs().amax().float() / _FP8_MAX).clamp(min=1e-12)
        st["w8"] = (weight.detach() / sb).to(torch.float8_e4m3fn).t()
        st["sb"] = sb
        st["wver"] = wver
    amax_now = x2d.abs().amax().float()
    sa = st.get("sa")
    if sa is None:
        sa = (amax_now / _FP8_MAX).clamp(min=1e-12)
    a8 = (x2d / sa).to(torch.float8_e4m3fn)
    st["sa"] = (amax_now / _FP8_MAX).clamp(min=1e-12)
    return torch._scaled_mm(
        a8, st["w8"], scale_a=sa, scale_b=st["sb"], bias=bias,
        out_dtype=torch.bfloat16,
    )


class _MaskColsInplace(torch.autograd.Function):
    """-inf fill of out[:, limit:] IN-PLACE with mark_dirty.

    Must run on the NON-VIEW 2-D output of LinearFunction, before any
    reshape: an in-place op on a view makes autograd rebase it as
    CopySlices, whose backward clones the full logits tensor (measured
    +6.8 ms/step at GPT-2 bench shape — profiles/ vocab-pad diff).  On
    the direct Function output it is a plain in-place node: no copies,
    pass-through gradient (downstream of -inf columns every consumer
    computes exp(-inf)=0, so their grad is exactly zero already)."""

    @staticmethod
    def forward(ctx, out, limit):
        out[:, limit:] = float("-inf")
        ctx.mark_dirty(out)
        return out

    @staticmethod
    def backward(ctx, grad):
        return grad, None


class LinearFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, activation, prefer_library=False, fp8=False):
        act = _ACT_MAP[activation]
        x2d = x.reshape(-1, x.shape[-1])
        ctx.x_shape = x.shape
        ctx.act = act
        pre_act = None
        if fp8 and _fp8_ok(x2d, weight):
            out = _fp8_linear(x2d.contiguous(), weight.contiguous(), bias)
            if act == _ACT_GELU:
                pre_act = out
                out = _gelu_tanh(out)
            elif act == _ACT_RELU:
                pre_act = out
                out = torch.relu(out)
            ctx.save_for_backward(
                x2d, weight, pre_act if pre_act is not None else torch.empty(0)
            )
            ctx.has_bias = bias is not None
            return out
        global _gelu_epi_broken
        if (
            _GELU_EPI
            and not _gelu_epi_broken
            and act == _ACT_GELU
            and bias is not None
            and _backend.use_native(x)
            and _backend.has_ext()
            and x.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16
            and hasattr(_backend.ext(), "gemm_bias_gelu_aux")
        ):
            try:
                out, pre_act = _backend.ext().gemm_bias_gelu_aux(
                    x2d.contiguous(), weight.contiguous(), bias.contiguous()
                )
                ctx.save_for_backward(x2d, weight, pre_act)
                ctx.has_bias = True
                return out
            except RuntimeError:
                _gelu_epi_broken = True  # no algo for this config: eager path
        m, n, k = x2d.shape[0], weight.shape[0], weight.shape[1]
        want_native = (
            _FWD_MODE == "custom"
            or (_FWD_MODE == "auto" and _custom_wins_shape(m, n, k))
        ) and not prefer_library
        if want_native and _backend.use_native(x) and _native_ok(x, weight):
            res = _backend.ext().gemm_nt(
                x2d.contiguous(), weight.contiguous(),
                bias if bias is not None else None, act,
                _custom_mode(m, n, k),
            )
            out = res[0]
            if act != _ACT_NONE:
                # epilogue stores the pre-activation too when an act is fused
                pre_act = res[1]
        else:
            out = torch.nn.functional.linear(x2d, weight, bias)
            if act == _ACT_GELU:
                pre_act = out
                out = _gelu_tanh(out)
            elif act == _ACT_RELU:
                pre_act = out
                out = torch.relu(out)
        ctx.save_for_backward(x2d, weight, pre_act if pre_act is not None else torch.empty(0))
        ctx.has_bias = bias is not None
        return out

    @staticmethod
    def backward(ctx, grad_out):
        x2d, weight, pre_act = ctx.saved_tensors
        g = grad_out.reshape(-1, grad_out.shape[-1])
        if ctx.act != _ACT_NONE:
            if _backend.use_native(g) and _backend.has_ext():
                g = _backend.ext().act_bwd(g, pre_act, ctx.act)  # fused single pass
            elif ctx.act == _ACT_GELU:
                pa = pre_act.float()
                c = 0.7978845608028654  # sqrt(2/pi): d/dx gelu_tanh(x)
                t = torch.tanh(c * (pa + 0.044715 * pa.pow(3)))
                dg = 0.5 * (1.0 + t) + 0.5 * pa * (1.0 - t * t) * c * (
                    1.0 + 3 * 0.044715 * pa.pow(2)
                )
                g = (g.float() * dg).to(g.dtype)
            elif ctx.act == _ACT_RELU:
                g = g * (pre_act > 0).to(g.dtype)
        grad_x = grad_w = grad_b = None
        if ctx.needs_input_grad[0]:
            grad_x = (g @ weight).reshape(ctx.x_shape)
        if ctx.needs_input_grad[1]:
            if _DEFER_WGRADS:
                # zero-bubble mode: queue the dW GEMM; autograd gets None
                # (no accumulation) and flush_deferred_wgrads() adds the
                # contribution into weight.grad later
                _DEFERRED.append((g, x2d, weight))
            else:
                # hand-written split-K TN kernel for the shapes it wins
                # (hipBLASLt runs these deep-contraction small-output
                # shapes at 208-513 TF, profiles r04)
                grad_w = _wgrad(g, x2d)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            if _backend.use_native(g) and _backend.has_ext():
                grad_b = _backend.ext().colsum(g)
            else:
                grad_b = g.sum(dim=0)
        return grad_x, grad_w, grad_b, None, None, None


def linear(
    x: torch.Tensor,
    weight: torch.Tensor,
    bias: Optional[torch.Tensor] = None,
    activation: Optional[str] = None,
    prefer_library: bool = False,
    fp8: bool = False,
    logical_out: Optional[int] = None,
) -> torch.Tensor:
    """y = activation(x @ weight.T + bias), fused on gfx950.

    ``prefer_library=True`` sends a PLAIN (no bias, no activation) GEMM
    through hipBLASLt instead — per the MI355X design split, the vendor
    library serves plain GEMMs (e.g. the tied LM head) while fused ones
    run the hand-written MFMA kernel.  ``logical_out`` (padded-vocab
    models): -inf-fill output columns >= logical_out in place before
    the caller sees the tensor.  ``fp8=True`` (experimental) runs
    the FORWARD in OCP e4m3 with cached weight casts + delayed activation
    scaling (~1.5-1.8x raw GEMM throughput; pays off once K is large
    enough that the activation cast pass is small next to the GEMM —
    GPT-2-small shapes measure net-negative, see NOTES_ROUND2.md);
    backward stays bf16."""
    out = LinearFunction.apply(x, weight, bias, activation, prefer_library, fp8)
    if logical_out is not None and logical_out < weight.shape[0]:
        # padded-vocab logits: mask the pad columns BEFORE the reshape
        # (on the non-view 2-D output) — see _MaskColsInplace
        if out.requires_grad:
            out = _MaskColsInplace.apply(out, logical_out)
        else:
            out[:, logical_out:] = float("-inf")
    # reshape OUTSIDE the custom Function: keeps the output a plain
    # autograd view so callers may do in-place ops on it
    return out.reshape(*x.shape[:-1], weight.shape[0])
