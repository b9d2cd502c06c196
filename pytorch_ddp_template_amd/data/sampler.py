"""ShardedSampler — a from-scratch DistributedSampler equivalent.

Same contract the reference relies on (reference ddp.py:138-141, 213-214):

* epoch-seeded global shuffle, identical on every rank (``set_epoch`` must be
  called each epoch to reshuffle — exactly the reference's
  ``train_sampler.set_epoch(epoch)``),
* rank r takes ``indices[r::world_size]`` of the shuffled order,
* pads with wrapped-around indices so every rank yields the same count
  (keeps collectives aligned across ranks).
"""

from __future__ import annotations

import math

import torch
from torch.utils.data import Sampler

from ..utils.dist import get_rank, get_world_size


class ShardedSampler(Sampler):
    def __init__(
        self,
        dataset,
        num_replicas: int | None = None,
        rank: int | None = None,
        shuffle: bool = True,
        seed: int = 0,
        drop_last: bool = False,
    ):
        self.dataset = dataset
        self.num_replicas = num_replicas if num_replicas is not None else get_world_size()
        self.rank = rank if rank is not None else get_rank()
        if not (0 <= self.rank < self.num_replicas):
            raise ValueError(
                f"rank {self.rank} out of range for world {self.num_replicas}"
            )
        self.shuffle = shuffle
        self.seed = seed
        self.drop_last = drop_last
        self.epoch = 0
        n = len(dataset)
        if drop_last:
            self.num_samples = n // self.num_replicas
        else:
            self.num_samples = math.ceil(n / self.num_replicas)
        self.total_size = self.num_samples * self.num_replicas

    def set_epoch(self, epoch: int) -> None:
        self.epoch = epoch

    def __iter__(self):
        n = len(self.dataset)
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed + self.epoch)
            indices = torch.randperm(n, generator=g).tolist()
        else:
            indices = list(range(n))
        if self.drop_last:
            indices = indices[: self.total_size]
        else:
            pad = self.total_size - len(indices)
            if pad > 0:
                indices += indices[:pad]
        return iter(indices[self.rank :: self.num_replicas])

    def __len__(self):
        return self.num_samples
