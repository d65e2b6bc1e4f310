"""Deterministic per-shard RNG streams.

The reference draws one independent RNG state per block from a master seed
(dask/array/random.py:75, dask/utils.py:382), making block results
independent of execution order. The MI355X equivalent: a counter-based seed
derived from (global_seed, shard_id, stream_id) via a splitmix64 hash feeding
a torch.Generator per shard. Bit-parity with the reference's Mersenne streams
is not required — only determinism and distributional parity.
"""

from __future__ import annotations

import torch


def _splitmix64(x: int) -> int:
    x = (x + 0x9E3779B97F4A7C15) & 0xFFFFFFFFFFFFFFFF
    z = x
    z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & 0xFFFFFFFFFFFFFFFF
    z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & 0xFFFFFFFFFFFFFFFF
    return z ^ (z >> 31)


def shard_seed(seed: int, shard_id: int, stream: int = 0) -> int:
    """Mix (seed, shard, stream) into one 63-bit generator seed."""
    h = _splitmix64(seed & 0xFFFFFFFFFFFFFFFF)
    h = _splitmix64(h ^ (shard_id * 0xD1342543DE82EF95))
    h = _splitmix64(h ^ (stream * 0x2545F4914F6CDD1D))
    return h & 0x7FFFFFFFFFFFFFFF


def generator(seed: int, shard_id: int, stream: int = 0,
              device: torch.device = None) -> torch.Generator:
    g = torch.Generator(device=device if device is not None else "cpu")
    g.manual_seed(shard_seed(seed, shard_id, stream))
    return g
