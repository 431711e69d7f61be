"""Shared helpers for the example launchers."""

from __future__ import annotations

import argparse

import torch
from torch.utils.data import DataLoader

from quintnet_amd.models import Model
from quintnet_amd.utils.data import CustomDataset, SyntheticMNIST


def parse_args(default_config: str = "examples/config.yaml"):
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default=default_config)
    ap.add_argument("--synthetic", action="store_true", default=False)
    ap.add_argument("--n-train", type=int, default=2048)
    ap.add_argument("--n-val", type=int, default=256)
    return ap.parse_args()


def build_model(cfg) -> Model:
    return Model(
        img_size=cfg.get("img_size", 28),
        patch_size=cfg.get("patch_size", 4),
        in_channels=cfg.get("in_channels", 1),
        hidden_dim=cfg.get("hidden_dim", 64),
        n_heads=cfg.get("n_heads", 4),
        depth=cfg.get("depth", 8),
        n_classes=cfg.get("n_classes", 10),
    )


def build_loaders(cfg, args, pg=None):
    """Train/val loaders; with a process-group manager whose dp axis is
    >1, each DP replica gets its OWN data shard via DistributedSampler
    (reference examples/full_3d.py:129 parity — without it every
    replica trains on identical batches and DP adds no data throughput)."""
    if cfg.get("dataset_path"):
        train = CustomDataset(cfg["dataset_path"], "train")
        val = CustomDataset(cfg["dataset_path"], "test")
    else:
        train = SyntheticMNIST(n=args.n_train, seed=0)
        val = SyntheticMNIST(n=args.n_val, seed=1)
    bs = cfg.get("batch_size", 8)
    nw = cfg.get("num_workers", 0)
    tr_sampler = va_sampler = None
    if pg is not None and getattr(pg, "dp_size", 1) > 1:
        from torch.utils.data.distributed import DistributedSampler

        tr_sampler = DistributedSampler(
            train, num_replicas=pg.dp_size, rank=pg.dp_rank,
            shuffle=True, seed=42,
        )
        va_sampler = DistributedSampler(
            val, num_replicas=pg.dp_size, rank=pg.dp_rank, shuffle=False,
        )
    return (
        DataLoader(train, batch_size=bs, num_workers=nw,
                   shuffle=False, sampler=tr_sampler),
        DataLoader(val, batch_size=bs, num_workers=nw,
                   shuffle=False, sampler=va_sampler),
    )


def device_type() -> str:
    return "cuda" if torch.cuda.is_available() else "cpu"
