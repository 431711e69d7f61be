"""Datasets & loaders (reference utils/Dataloader.py parity).

CustomDataset / mnist_transform for the ViT path;
SummarizationDataset/Collator/DataLoader for the GPT-2 path; plus
synthetic dataset builders (no network → benchmarks use synthetic data
of the real shapes with random-init weights).
"""

from __future__ import annotations

from typing import Optional

import torch
from torch.utils.data import DataLoader, Dataset

__all__ = [
    "CustomDataset",
    "mnist_transform",
    "SyntheticMNIST",
    "SyntheticCLM",
    "SummarizationDataset",
    "SummarizationCollator",
    "SummarizationDataLoader",
]

_MNIST_MEAN, _MNIST_STD = 0.1307, 0.3081


def mnist_transform(img):
    """ToTensor + Normalize(0.1307, 0.3081) without torchvision."""
    import numpy as np

    if isinstance(img, torch.Tensor):
        x = img.float()
    else:
        x = torch.from_numpy(np.array(img, dtype="float32") / 255.0)
    if x.dim() == 2:
        x = x.unsqueeze(0)
    return (x - _MNIST_MEAN) / _MNIST_STD


class CustomDataset(Dataset):
    """HF-datasets-on-disk MNIST-style dataset ({'image','label'} dicts)."""

    def __init__(self, dataset_path: str, split: str = "train", transform=mnist_transform):
        from datasets import load_from_disk

        ds = load_from_disk(dataset_path)
        try:
            self.data = ds[split]
        except (KeyError, TypeError):
            self.data = ds
        self.transform = transform

    def __len__(self):
        return len(self.data)

    def __getitem__(self, idx):
        row = self.data[idx]
        img = self.transform(row["image"]) if self.transform else row["image"]
        return {"images": img, "labels": int(row["label"])}


class SyntheticMNIST(Dataset):
    """Random MNIST-shaped data (28×28×1, 10 classes) for benchmarks."""

    def __init__(self, n: int = 2048, seed: int = 0, dtype=torch.float32):
        g = torch.Generator().manual_seed(seed)
        self.x = torch.randn(n, 1, 28, 28, generator=g, dtype=torch.float32).to(dtype)
        self.y = torch.randint(0, 10, (n,), generator=g)

    def __len__(self):
        return len(self.y)

    def __getitem__(self, i):
        return {"images": self.x[i], "labels": self.y[i]}


class SyntheticCLM(Dataset):
    """Random token sequences for causal-LM benchmarks."""

    def __init__(self, n: int = 512, seq_len: int = 1024, vocab_size: int = 50257, seed: int = 0):
        g = torch.Generator().manual_seed(seed)
        self.ids = torch.randint(0, vocab_size, (n, seq_len), generator=g)

    def __len__(self):
        return self.ids.shape[0]

    def __getitem__(self, i):
        ids = self.ids[i]
        return {"input_ids": ids, "labels": ids.clone()}


class SummarizationDataset(Dataset):
    """CSV with 'article' and 'highlights' columns (reference :216-260)."""

    def __init__(self, csv_path: str, max_samples: Optional[int] = None):
        import pandas as pd

        df = pd.read_csv(csv_path)
        if max_samples:
            df = df.iloc[:max_samples]
        self.articles = df["article"].tolist()
        self.highlights = df["highlights"].tolist()

    def __len__(self):
        return len(self.articles)

    def __getitem__(self, i):
        return {"article": self.articles[i], "highlights": self.highlights[i]}


class SummarizationCollator:
    """'article\\n\\nTL;DR: highlights<eos>', padded to max_length,
    labels = -100 on padding/prompt (reference :263-319)."""

    def __init__(self, tokenizer, max_length: int = 512):
        self.tok = tokenizer
        self.max_length = max_length
        if self.tok.pad_token is None:
            self.tok.pad_token = self.tok.eos_token

    def __call__(self, batch):
        texts = [
            f"{b['article']}\n\nTL;DR: {b['highlights']}{self.tok.eos_token}" for b in batch
        ]
        enc = self.tok(
            texts,
            max_length=self.max_length,
            truncation=True,
            padding="max_length",
            return_tensors="pt",
        )
        labels = enc["input_ids"].clone()
        labels[enc["attention_mask"] == 0] = -100
        return {
            "input_ids": enc["input_ids"],
            "attention_mask": enc["attention_mask"],
            "labels": labels,
        }


def SummarizationDataLoader(dataset, tokenizer, batch_size=8, max_length=512,
                            shuffle=True, dp_rank: int = 0, dp_size: int = 1,
                            **kw):
    """Collated loader for the summarization set; with dp_size > 1 each
    DP replica reads its own shard (DistributedSampler — matches the
    reference's full_3d.py:129 DP data sharding)."""
    sampler = None
    if dp_size > 1:
        from torch.utils.data.distributed import DistributedSampler

        sampler = DistributedSampler(dataset, num_replicas=dp_size,
                                     rank=dp_rank, shuffle=shuffle, seed=42)
        shuffle = False
    return DataLoader(
        dataset,
        batch_size=batch_size,
        shuffle=shuffle,
        sampler=sampler,
        collate_fn=SummarizationCollator(tokenizer, max_length),
        **kw,
    )
