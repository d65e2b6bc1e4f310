from . import comm, coordination, futures

__all__ = ["comm", "coordination", "futures"]
