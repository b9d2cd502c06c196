from .distributed import DistributedModel

__all__ = ["DistributedModel"]
