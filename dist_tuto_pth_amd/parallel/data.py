"""Data partitioning layer (L4 of SURVEY.md §1; train_dist.py:17-50,74-91).

``Partition`` and ``DataPartitioner`` reproduce the reference semantics
exactly: a deterministic seeded shuffle shared by every rank yields
disjoint per-rank shards (same seed => identical permutation,
train_dist.py:35-47).

``partition_dataset`` keeps the reference's shape (equal ``1.0/size``
fractions, per-rank batch ``batch // world_size``, shuffling loader —
train_dist.py:74-91) but draws from a synthetic MNIST-shaped dataset:
this environment has no network, and BASELINE.md config 3 is defined on
synthetic 28x28 data with random-init weights.
"""

from __future__ import annotations

from random import Random
from typing import List, Sequence

import torch
from torch.utils.data import DataLoader, Dataset

from .. import dist


class Partition(Dataset):
    """Read-only index-remapped view of a dataset (train_dist.py:17-29)."""

    def __init__(self, data, index: Sequence[int]):
        self.data = data
        self.index = list(index)

    def __len__(self):
        return len(self.index)

    def __getitem__(self, i):
        return self.data[self.index[i]]


class DataPartitioner:
    """Seeded shuffle + consecutive fractional slices
    (train_dist.py:32-50).  Identical seed on every rank gives disjoint
    shards."""

    def __init__(self, data, sizes=(0.7, 0.2, 0.1), seed: int = 1234):
        self.data = data
        self.partitions: List[List[int]] = []
        rng = Random()
        rng.seed(seed)
        indexes = list(range(len(data)))
        rng.shuffle(indexes)
        n = len(data)
        for frac in sizes:
            part_len = int(frac * n)
            self.partitions.append(indexes[:part_len])
            indexes = indexes[part_len:]

    def use(self, partition: int) -> Partition:
        return Partition(self.data, self.partitions[partition])


class SyntheticMNIST(Dataset):
    """MNIST-shaped synthetic data: normalized 1x28x28 floats + labels.

    Stands in for torchvision's MNIST download (train_dist.py:76-83) —
    deterministic per index so every rank sees the same dataset, like
    the shared download."""

    def __init__(self, n: int = 60000, seed: int = 1234,
                 device: str = "cpu"):
        g = torch.Generator().manual_seed(seed)
        # match the post-Normalize((0.1307,),(0.3081,)) statistics
        self.images = torch.randn(n, 1, 28, 28, generator=g)
        self.labels = torch.randint(0, 10, (n,), generator=g)

    def __len__(self):
        return self.images.shape[0]

    def __getitem__(self, i):
        return self.images[i], self.labels[i]


def partition_dataset(dataset=None, batch_size: int = 128,
                      seed: int = 1234, num_workers: int = 0):
    """Shard the dataset across ranks (train_dist.py:74-91).

    Uses integer division for the per-rank batch (the committed
    ``128 // size``, train_dist.py:85 — not the paper's float-division
    bug, tuto.md:267) and equal ``1/size`` shards."""
    if dataset is None:
        dataset = SyntheticMNIST(seed=seed)
    size = dist.get_world_size()
    bsz = batch_size // size
    partition_sizes = [1.0 / size for _ in range(size)]
    partition = DataPartitioner(dataset, partition_sizes, seed=seed)
    shard = partition.use(dist.get_rank())
    train_set = DataLoader(shard, batch_size=bsz, shuffle=True,
                           num_workers=num_workers)
    return train_set, bsz
