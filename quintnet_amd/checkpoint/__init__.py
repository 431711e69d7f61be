"""Checkpointing: per-rank shards, offline merge, staged distributed load.

Shard layout parity with the reference (GPT2_Trainer.py:453-507):
``{out_dir}/{name}_pp{p}_tp{t}.pt`` with ``model_state_dict``,
``optimizer_state_dict`` and ``parallelism_info`` metadata; the merge
CLI (merge.py) recombines TP (cat) and PP (key remap) shards into HF
GPT-2 format.
"""

from __future__ import annotations

import os
from typing import Any, Dict, Optional

import torch
import torch.distributed as dist

from .merge import merge_checkpoints
from .distributed_loading import load_gpt2_distributed

__all__ = [
    "save_checkpoint",
    "load_checkpoint",
    "save_sharded_checkpoint",
    "load_sharded_checkpoint",
    "merge_checkpoints",
    "load_gpt2_distributed",
]


def _extra_axes_suffix(pg_manager) -> str:
    """Filename suffix for model-sharding axes beyond pp/tp (cp, ep, ...)
    so their shards don't collide; empty for plain [dp, tp, pp] meshes
    (keeps the reference-compatible naming)."""
    if pg_manager is None:
        return ""
    out = ""
    for ax in getattr(pg_manager, "mesh_name", ()):
        if ax in ("dp", "tp", "pp"):
            continue
        out += f"_{ax}{pg_manager.axis_rank(ax)}"
    return out


def save_sharded_checkpoint(
    model,
    out_dir: str,
    name: str = "final_model",
    pg_manager=None,
    optimizer=None,
    config: Optional[Dict[str, Any]] = None,
) -> Optional[str]:
    """Write this rank's shard ``{name}_pp{p}_tp{t}[{extra}].pt``.

    ``extra`` covers any further model-sharding mesh axes (cp, ep, ...):
    without it, MoE/CP ranks with equal (pp, tp) would clobber each
    other's shards.  dp rank 0 of each shard coordinate writes.
    """
    for m in model.modules():
        if type(m).__name__ == "ZeRO3Block":
            raise ValueError(
                "save_sharded_checkpoint: model is ZeRO-3 wrapped — its "
                "state dict holds flat shards, not the named-parameter "
                "contract the merge CLI expects.  Re-gather with "
                "ZeRO3Block.full_state_dict_tensors() per block (see "
                "parallel/zero3.py) before saving."
            )
    pp_rank = pg_manager.pp_rank if pg_manager is not None else 0
    tp_rank = pg_manager.tp_rank if pg_manager is not None else 0
    dp_rank = pg_manager.dp_rank if pg_manager is not None else 0
    pp_size = pg_manager.pp_size if pg_manager is not None else 1
    tp_size = pg_manager.tp_size if pg_manager is not None else 1
    extra = _extra_axes_suffix(pg_manager)

    os.makedirs(out_dir, exist_ok=True)
    path = None
    if dp_rank == 0:
        path = os.path.join(out_dir, f"{name}_pp{pp_rank}_tp{tp_rank}{extra}.pt")
        info = {
            "pp_rank": pp_rank,
            "pp_size": pp_size,
            "tp_rank": tp_rank,
            "tp_size": tp_size,
            "dp_rank": dp_rank,
        }
        # interleaved (virtual-pipeline) wrapper: record the chunk→global-
        # stage map so the merge CLI can rename chunk-local parameter
        # names (chunks.c.i.*) to global layers
        if hasattr(model, "chunks") and hasattr(model, "layer_distribution"):
            info["interleaved"] = {
                "num_chunks": int(model.num_chunks),
                "pp_size": int(model.pp_size),
                "layer_distribution": [list(g) for g in model.layer_distribution],
            }
        payload = {
            "model_state_dict": {k: v.cpu() for k, v in model.state_dict().items()},
            "parallelism_info": info,
            "config": config or {},
        }
        torch.save(payload, path)
    # ZeRO-1 optimizer state is sharded over the DP axis too: every dp
    # rank writes its own shard (mid-training resume — a capability the
    # reference never finished).
    if optimizer is not None and hasattr(optimizer, "state_dict"):
        opath = os.path.join(
            out_dir, f"{name}_optim_pp{pp_rank}_tp{tp_rank}{extra}_dp{dp_rank}.pt"
        )
        try:
            sd = optimizer.state_dict()
            sd = {
                k: (v.cpu() if isinstance(v, torch.Tensor) else v) for k, v in sd.items()
            }
            torch.save(sd, opath)
        except Exception as e:  # noqa: BLE001 — model shards must still land
            import warnings

            rank = dist.get_rank() if dist.is_initialized() else 0
            warnings.warn(
                f"[rank {rank}] optimizer shard save FAILED at {opath}: {e!r} — "
                "a later resume will restore weights with stale/missing "
                "optimizer state"
            )
    if dist.is_initialized():
        dist.barrier()
    return path


def load_sharded_checkpoint(
    model, out_dir: str, name: str, pg_manager=None, optimizer=None, strict=True
):
    """Reload this rank's model shard (and, for mid-training resume, its
    ZeRO optimizer shard when ``optimizer`` is given)."""
    for m in model.modules():
        if type(m).__name__ == "ZeRO3Block":
            raise ValueError(
                "save_sharded_checkpoint: model is ZeRO-3 wrapped — its "
                "state dict holds flat shards, not the named-parameter "
                "contract the merge CLI expects.  Re-gather with "
                "ZeRO3Block.full_state_dict_tensors() per block (see "
                "parallel/zero3.py) before saving."
            )
    pp_rank = pg_manager.pp_rank if pg_manager is not None else 0
    tp_rank = pg_manager.tp_rank if pg_manager is not None else 0
    dp_rank = pg_manager.dp_rank if pg_manager is not None else 0
    extra = _extra_axes_suffix(pg_manager)
    path = os.path.join(out_dir, f"{name}_pp{pp_rank}_tp{tp_rank}{extra}.pt")
    ckpt = torch.load(path, map_location="cpu", weights_only=False)
    model.load_state_dict(ckpt["model_state_dict"], strict=strict)
    if optimizer is not None:
        if hasattr(optimizer, "refresh_master_"):
            optimizer.refresh_master_()
        opath = os.path.join(
            out_dir, f"{name}_optim_pp{pp_rank}_tp{tp_rank}{extra}_dp{dp_rank}.pt"
        )
        if os.path.exists(opath):
            osd = torch.load(opath, map_location="cpu", weights_only=False)
            dev = next(model.parameters()).device
            osd = {
                k: (v.to(dev) if isinstance(v, torch.Tensor) else v)
                for k, v in osd.items()
            }
            optimizer.load_state_dict(osd)
        else:
            import warnings

            warnings.warn(
                f"optimizer shard missing at {opath}; resuming with FRESH "
                "optimizer state (weights restored, Adam moments reset)"
            )
    return ckpt


def save_checkpoint(model, path: str, optimizer=None, **extra) -> str:
    """Single-file save (reference utils/utils.py:54-74 parity): model
    state dict (+ optional optimizer state and any extra metadata)."""
    payload: Dict[str, Any] = {"model_state_dict": model.state_dict(), **extra}
    if optimizer is not None:
        payload["optimizer_state_dict"] = optimizer.state_dict()
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    torch.save(payload, path)
    return path


def load_checkpoint(model, path: str, optimizer=None, strict: bool = True):
    """Single-file load (reference utils/utils.py:77-96 parity)."""
    ckpt = torch.load(path, map_location="cpu", weights_only=False)
    model.load_state_dict(ckpt.get("model_state_dict", ckpt), strict=strict)
    if optimizer is not None and "optimizer_state_dict" in ckpt:
        optimizer.load_state_dict(ckpt["optimizer_state_dict"])
    return ckpt
