"""Build/feature diagnostics: ``python -m quintnet_amd.info``.

Prints what a bug report needs: torch/ROCm versions, whether the gfx950
extension is loaded, and the state of every performance-relevant env
knob."""

from __future__ import annotations

import os

KNOBS = [
    ("QN_GEMM_FWD", "auto", "forward GEMM dispatch (auto|custom|library)"),
    ("QN_WGRAD", "auto", "wgrad dispatch (auto|custom|library)"),
    ("QN_WGRAD_SPLITS", "(sweep)", "split-K override"),
    ("QN_WGRAD_LDS", "64", "32 = 32KB-stage wgrad variant (4 waves)"),
    ("QN_ATTN_FWD_OCC", "3", "attention fwd waves/SIMD template"),
    ("QN_ATTN_KLDS", "1", "cooperative K-through-LDS staging"),
    ("QN_ATTN_DQ_OCC", "3", "4 = non-pipelined 4-wave dq variant"),
    ("QN_ATTN_DKV_OCC", "2", "3 = non-pipelined 3-wave dkv variant"),
    ("QN_GELU_EPI", "0", "hipBLASLt GELU_AUX_BIAS epilogue (exp.)"),
    ("QN_GRAPHS", "1", "whole-step hipGraph capture in bench"),
    ("QN_VOCAB_PAD", "128/0", "padded-vocab logits width (bench)"),
    ("QN_ACT_CKPT", "0", "activation checkpointing in bench"),
    ("QN_CP_RING", "-", "context-parallel ring flavor"),
    ("QN_SCHEDULE", "1f1b", "pipeline schedule override (bench)"),
    ("QN_WATCHDOG", "1", "bench hang watchdog at world>1"),
    ("PYTORCH_TUNABLEOP_TUNING", "auto", "hipBLASLt algo tuning"),
]


def report() -> str:
    import torch

    from . import ops

    lines = [
        f"torch            : {torch.__version__}",
        f"hip              : {getattr(torch.version, 'hip', None)}",
        f"cuda_available   : {torch.cuda.is_available()}",
    ]
    if torch.cuda.is_available():
        lines.append(f"device           : {torch.cuda.get_device_name(0)}")
    has = ops.has_ext()
    lines.append(f"native extension : {'loaded' if has else 'MISSING'}")
    if has:
        lines.append(f"ext arch         : {ops.ext().gfx}")
        lines.append(
            "gelu_epilogue    : "
            + ("available" if hasattr(ops.ext(), "gemm_bias_gelu_aux") else "absent")
        )
    lines.append("")
    lines.append(f"{'env knob':26s} {'value':12s} default/meaning")
    for k, d, m in KNOBS:
        v = os.environ.get(k, "(unset)")
        lines.append(f"{k:26s} {v:12s} [{d}] {m}")
    return "\n".join(lines)


if __name__ == "__main__":
    print(report())
