"""Data utilities for the examples: synthetic datasets (no network in
this environment) and the IID / non-IID worker sharding samplers.

Parity targets (examples/utils.py):
  - SplitSampler: each worker trains on its contiguous 1/W shard (IID)
  - ClassSplitSampler: each worker gets a disjoint subset of classes
    (the non-IID / label-skew geo-distributed setting)
"""

from __future__ import annotations

from typing import Iterator, List, Optional

import torch
from torch.utils.data import Dataset, Sampler


class SyntheticImageDataset(Dataset):
    """Deterministic synthetic labelled images (replaces the MNIST /
    FMNIST / CIFAR10 loaders; there is no dataset download here)."""

    def __init__(self, n: int = 2048, shape=(3, 224, 224), num_classes=10,
                 seed: int = 0, proto_seed: int = 1234):
        g = torch.Generator().manual_seed(seed)
        self.x = torch.randn(n, *shape, generator=g)
        self.y = torch.randint(0, num_classes, (n,), generator=g)
        # learnable task: bias each image by a fixed class prototype drawn
        # from proto_seed, SHARED between train and test splits
        pg = torch.Generator().manual_seed(proto_seed)
        for c in range(num_classes):
            proto = torch.randn(*shape, generator=pg)
            self.x[self.y == c] += 0.75 * proto

    def __len__(self):
        return self.x.shape[0]

    def __getitem__(self, i):
        return self.x[i], self.y[i]


class SplitSampler(Sampler):
    """Contiguous 1/num_parts shard for worker `part_index` (IID)."""

    def __init__(self, length: int, num_parts: int, part_index: int,
                 shuffle: bool = True, seed: int = 0):
        self.length = length
        self.num_parts = num_parts
        self.part_index = part_index
        self.shuffle = shuffle
        self.seed = seed
        self.epoch = 0
        per = length // num_parts
        self.lo = per * part_index
        self.hi = per * (part_index + 1) if part_index < num_parts - 1 else length

    def set_epoch(self, epoch: int):
        self.epoch = epoch

    def __iter__(self) -> Iterator[int]:
        idx = list(range(self.lo, self.hi))
        if self.shuffle:
            g = torch.Generator().manual_seed(self.seed + self.epoch)
            perm = torch.randperm(len(idx), generator=g)
            idx = [idx[i] for i in perm]
        return iter(idx)

    def __len__(self):
        return self.hi - self.lo


class ClassSplitSampler(Sampler):
    """Non-IID: worker `part_index` sees only its share of the classes."""

    def __init__(self, labels: torch.Tensor, num_parts: int, part_index: int,
                 num_classes: Optional[int] = None, shuffle: bool = True,
                 seed: int = 0):
        num_classes = num_classes or int(labels.max().item()) + 1
        classes = [c for c in range(num_classes) if c % num_parts == part_index]
        mask = torch.zeros(labels.numel(), dtype=torch.bool)
        for c in classes:
            mask |= labels == c
        self.indices: List[int] = mask.nonzero().flatten().tolist()
        self.shuffle = shuffle
        self.seed = seed
        self.epoch = 0

    def set_epoch(self, epoch: int):
        self.epoch = epoch

    def __iter__(self):
        idx = list(self.indices)
        if self.shuffle:
            g = torch.Generator().manual_seed(self.seed + self.epoch)
            perm = torch.randperm(len(idx), generator=g)
            idx = [idx[i] for i in perm]
        return iter(idx)

    def __len__(self):
        return len(self.indices)


def worker_loader(dataset: Dataset, batch_size: int, num_workers_total: int,
                  worker_index: int, split_by_class: bool = False,
                  seed: int = 0):
    if split_by_class:
        sampler = ClassSplitSampler(dataset.y, num_workers_total,
                                    worker_index, seed=seed)
    else:
        sampler = SplitSampler(len(dataset), num_workers_total, worker_index,
                               seed=seed)
    return torch.utils.data.DataLoader(dataset, batch_size=batch_size,
                                       sampler=sampler, drop_last=True)
