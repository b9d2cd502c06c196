"""Rank helpers with safe single-process defaults (reference utils.py:84-101)."""

from __future__ import annotations

import torch.distributed as dist


def get_rank() -> int:
    """Global rank; 0 when torch.distributed is unavailable/uninitialized
    (reference utils.py:84-92)."""
    if dist.is_available() and dist.is_initialized():
        return dist.get_rank()
    return 0


def get_world_size() -> int:
    """World size; 1 when torch.distributed is unavailable/uninitialized
    (reference utils.py:94-99)."""
    if dist.is_available() and dist.is_initialized():
        return dist.get_world_size()
    return 1


def is_main_process() -> bool:
    """rank == 0 (reference utils.py:101)."""
    return get_rank() == 0
