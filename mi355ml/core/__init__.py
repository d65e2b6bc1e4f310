from .array import DistArray, from_numpy, shard_rows
from . import random

__all__ = ["DistArray", "from_numpy", "shard_rows", "random"]
