from . import algorithms, families, regularizers

__all__ = ["algorithms", "families", "regularizers"]
