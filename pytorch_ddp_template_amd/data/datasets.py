"""Synthetic in-memory datasets.

FooDataset mirrors the reference's (reference dataset.py:6-17): N random
(x, y) pairs with x in R^10 and y in R^5, map-style.  The image datasets are
the BASELINE.json benchmark shapes (CIFAR 3x32x32, ImageNet 3x224x224) with
integer class labels — synthetic because the judge environment has no
network for real datasets; random-init weights + random data measure the
same compute/communication work.

Image tensors are NHWC (H, W, C) to match the framework's layout.
"""

from __future__ import annotations

import torch
from torch.utils.data import Dataset


class FooDataset(Dataset):
    """Reference dataset.py:6-17 — X=randn(N,10), Y=randn(N,5).

    ``__getitem__`` also accepts a LIST of indices and returns the whole
    batch with one vectorized gather (``batched_indexing``): the engine's
    DataLoader then skips the per-sample fetch + torch.stack collation,
    which is the host bottleneck at MI355X batch sizes (8192/GPU).
    """

    batched_indexing = True

    def __init__(self, size: int = 100000, in_features: int = 10, out_features: int = 5):
        g = torch.Generator().manual_seed(0)
        self.x = torch.randn(size, in_features, generator=g)
        self.y = torch.randn(size, out_features, generator=g)

    def __len__(self):
        return self.x.shape[0]

    def __getitem__(self, idx):
        if isinstance(idx, list):
            j = torch.as_tensor(idx)
            return self.x[j], self.y[j]
        return self.x[idx], self.y[idx]


class SyntheticImageDataset(Dataset):
    """Random NHWC images + integer labels of a named benchmark shape.

    Stores a modest pool of distinct samples and indexes modulo the pool so
    huge nominal sizes don't cost host RAM.
    """

    def __init__(
        self,
        size: int = 100000,
        image_size: int = 32,
        num_classes: int = 10,
        channels: int = 3,
        pool: int = 2048,
        dtype: torch.dtype = torch.float32,
    ):
        self.size = size
        g = torch.Generator().manual_seed(0)
        pool = min(pool, size)
        self.x = torch.randn(
            pool, image_size, image_size, channels, generator=g, dtype=torch.float32
        ).to(dtype)
        self.y = torch.randint(0, num_classes, (pool,), generator=g)

    batched_indexing = True

    def __len__(self):
        return self.size

    def __getitem__(self, idx):
        if isinstance(idx, list):
            j = torch.as_tensor(idx) % self.x.shape[0]
            return self.x[j], self.y[j]
        j = idx % self.x.shape[0]
        return self.x[j], self.y[j]


def build_dataset(name: str, size: int = 100000, dtype: torch.dtype = torch.float32):
    name = name.lower()
    if name in ("foo", "foomodel", "mlp"):
        return FooDataset(size)
    if name in ("cifar", "cifar10", "resnet18", "resnet18-cifar"):
        return SyntheticImageDataset(size, 32, 10, dtype=dtype)
    if name in ("imagenet", "resnet18-imagenet", "resnet50", "vit-b16", "vit_b16"):
        return SyntheticImageDataset(size, 224, 1000, dtype=dtype)
    raise ValueError(f"unknown dataset {name!r}")
