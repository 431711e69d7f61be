"""Fused cross-entropy (log-softmax + NLL) with ignore_index.

One hand-written CDNA4 kernel pass per direction over the [N, V] logits
(V = 50257 for GPT-2): fwd computes per-row max/logsumexp/nll without
materializing log-probs; bwd writes (softmax - onehot)/n_valid in one
pass.  Replaces reference nn.CrossEntropyLoss (trainer.py:90,
GPT2_Trainer.py:109).
"""

from __future__ import annotations

import torch

from . import _backend

__all__ = ["cross_entropy", "CrossEntropyFunction"]


class CrossEntropyFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, target, ignore_index):
        # logits: [N, V] (any float dtype), target: [N] int64
        if _backend.use_native(logits) and _backend.has_ext():
            loss_sum, n_valid, lse = _backend.ext().cross_entropy_fwd(
                logits.contiguous(), target.contiguous(), ignore_index
            )
        else:
            lf = logits.float()
            lse = torch.logsumexp(lf, dim=-1)
            valid = target != ignore_index
            n_valid = valid.sum()
            tgt = target.clamp_min(0)
            nll = lse - lf.gather(1, tgt.unsqueeze(1)).squeeze(1)
            nll = torch.where(valid, nll, torch.zeros_like(nll))
            loss_sum = nll.sum()
        ctx.save_for_backward(logits, target, lse, n_valid)
        ctx.ignore_index = ignore_index
        n = n_valid.clamp_min(1).to(loss_sum.dtype)
        return loss_sum / n

    @staticmethod
    def backward(ctx, grad_output):
        logits, target, lse, n_valid = ctx.saved_tensors
        if _backend.use_native(logits) and _backend.has_ext():
            # grad scale fused into the kernel (no extra 800 MB pass)
            dl = _backend.ext().cross_entropy_bwd(
                logits.contiguous(), target.contiguous(), lse, n_valid,
                ctx.ignore_index, grad_output,
            )
        else:
            lf = logits.float()
            p = torch.exp(lf - lse.unsqueeze(1))
            valid = (target != ctx.ignore_index).unsqueeze(1)
            tgt = target.clamp_min(0)
            p.scatter_add_(
                1, tgt.unsqueeze(1), torch.full_like(tgt.unsqueeze(1), -1.0, dtype=p.dtype)
            )
            n = n_valid.clamp_min(1).float()
            dl = torch.where(valid, p / n, torch.zeros_like(p)) * grad_output
            dl = dl.to(logits.dtype)
        return dl, None, None


def cross_entropy(
    logits: torch.Tensor, target: torch.Tensor, ignore_index: int = -100,
    label_smoothing: float = 0.0,
) -> torch.Tensor:
    """Mean cross-entropy over target != ignore_index rows.

    ``label_smoothing`` > 0 uses a torch-composed implementation (the
    smoothed loss needs a per-row mean-logit term the fused kernel does
    not produce yet — r3 candidate: one extra accumulator in the same
    pass); 0 keeps the single-pass HIP kernel."""
    l2 = logits.reshape(-1, logits.shape[-1])
    t1 = target.reshape(-1)
    if label_smoothing and label_smoothing > 0.0:
        lf = l2.float()
        lse = torch.logsumexp(lf, dim=-1)
        valid = t1 != ignore_index
        n = valid.sum().clamp_min(1)
        tgt = t1.clamp_min(0)
        nll = lse - lf.gather(1, tgt.unsqueeze(1)).squeeze(1)
        # uniform-smoothed term: mean_j (lse - logit_j)
        smooth = lse - lf.mean(dim=-1)
        per_row = (1.0 - label_smoothing) * nll + label_smoothing * smooth
        per_row = torch.where(valid, per_row, torch.zeros_like(per_row))
        return per_row.sum() / n.to(per_row.dtype)
    return CrossEntropyFunction.apply(l2, t1, ignore_index)


def shift_labels(labels: torch.Tensor, ignore_index: int = -100) -> torch.Tensor:
    """Next-token targets aligned with FULL-length logits: position t is
    labeled with token t+1, last position ignored.  Lets causal-LM loss
    run on the unsliced logits tensor — ``logits[:, :-1]`` forces an
    800 MB copy of the GPT-2 logits at reshape time."""
    tg = torch.full_like(labels, ignore_index)
    tg[:, :-1] = labels[:, 1:]
    return tg


def causal_lm_loss(
    logits: torch.Tensor, labels: torch.Tensor, ignore_index: int = -100,
    label_smoothing: float = 0.0,
) -> torch.Tensor:
    """Mean next-token cross-entropy on full-length logits/labels [B,T(,V)]."""
    return cross_entropy(logits, shift_labels(labels, ignore_index),
                         ignore_index, label_smoothing)
