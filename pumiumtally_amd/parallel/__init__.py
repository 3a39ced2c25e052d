from .dist import DistributedTally, init_distributed

__all__ = ["DistributedTally", "init_distributed"]
