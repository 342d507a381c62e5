from .distributed import DistributedDataParallel

__all__ = ["DistributedDataParallel"]
