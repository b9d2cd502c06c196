from .datasets import FooDataset, SyntheticImageDataset, build_dataset
from .prefetch import CudaPrefetcher
from .sampler import ShardedSampler

__all__ = [
    "FooDataset",
    "SyntheticImageDataset",
    "build_dataset",
    "ShardedSampler",
    "CudaPrefetcher",
]
