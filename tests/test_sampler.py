"""ShardedSampler contract (DistributedSampler reimplementation)."""

import pytest
import torch

from pytorch_ddp_template_amd.data import FooDataset, ShardedSampler


class _Sized:
    def __init__(self, n):
        self.n = n

    def __len__(self):
        return self.n


def all_shards(n, world, epoch=0, shuffle=True):
    shards = []
    for r in range(world):
        s = ShardedSampler(_Sized(n), num_replicas=world, rank=r, shuffle=shuffle)
        s.set_epoch(epoch)
        shards.append(list(iter(s)))
    return shards


def test_shards_partition_dataset():
    shards = all_shards(100, 4)
    flat = sorted(i for s in shards for i in s)
    assert flat == sorted(range(100))
    assert all(len(s) == 25 for s in shards)


def test_padding_when_not_divisible():
    shards = all_shards(10, 4)
    # every rank yields the same count; union covers the dataset
    assert len({len(s) for s in shards}) == 1
    assert set(range(10)) <= {i for s in shards for i in s}


def test_set_epoch_reshuffles():
    a = all_shards(1000, 2, epoch=0)
    b = all_shards(1000, 2, epoch=1)
    assert a != b
    # same epoch -> deterministic
    assert a == all_shards(1000, 2, epoch=0)


def test_no_shuffle_is_strided():
    shards = all_shards(12, 3, shuffle=False)
    assert shards[0] == [0, 3, 6, 9]
    assert shards[1] == [1, 4, 7, 10]


def test_rank_validation():
    with pytest.raises(ValueError):
        ShardedSampler(_Sized(10), num_replicas=2, rank=5)


def test_foo_dataset_shapes():
    ds = FooDataset(64)
    x, y = ds[0]
    assert x.shape == (10,) and y.shape == (5,)
    assert len(ds) == 64
    # deterministic
    x2, _ = FooDataset(64)[0]
    assert torch.equal(x, x2)
