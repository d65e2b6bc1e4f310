from . import comm, coordination

__all__ = ["comm", "coordination"]
