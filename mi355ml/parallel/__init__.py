from . import comm

__all__ = ["comm"]
