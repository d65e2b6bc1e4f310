"""Layered configuration: defaults < yaml file < MI355ML_* env vars < config.set().

MI355X-native analog of the reference's dask.config (dask/config.py:127,168,402):
a single flat-nested dict, `__`-nested env vars, and a context-manager `set()`.
No schema machinery — the config surface here is small and explicit.
"""

from __future__ import annotations

import contextlib
import copy
import os
import threading
from typing import Any, Iterator

import yaml

DEFAULTS: dict = {
    "comm": {
        # torch.distributed backend: "auto" -> nccl(=RCCL) if cuda else gloo
        "backend": "auto",
        # seconds before a collective times out (failure detection)
        "timeout": 300.0,
        # overlap collectives with compute on a side stream where supported
        "overlap": True,
    },
    "array": {
        # default row-block size for blocked/partial_fit iteration on a shard
        "block_rows": 1_000_000,
    },
    "kernels": {
        # require the HIP extension for CUDA tensors (fail loudly, never silently
        # fall back to eager torch on a GPU box)
        "require_hip": True,
    },
    "search": {
        # intra-rank thread parallelism for adaptive-search trials (the
        # reference runs one sklearn fit per dask worker THREAD; sklearn's
        # SGD releases the GIL). 0 = serial.
        "threads": 8,
    },
    "random": {"seed": 0},
    "logging": {"level": "WARNING"},
}

_local = threading.local()


def _deep_merge(base: dict, upd: dict) -> dict:
    out = copy.deepcopy(base)
    for k, v in upd.items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _deep_merge(out[k], v)
        else:
            out[k] = v
    return out


def _coerce(value: str) -> Any:
    try:
        return yaml.safe_load(value)
    except Exception:
        return value


def _collect_env(env: dict | None = None) -> dict:
    """MI355ML_FOO__BAR=1 -> {"foo": {"bar": 1}} (reference: dask/config.py:168)."""
    env = os.environ if env is None else env
    out: dict = {}
    for name, value in env.items():
        if not name.startswith("MI355ML_"):
            continue
        keys = name[len("MI355ML_"):].lower().split("__")
        d = out
        for k in keys[:-1]:
            d = d.setdefault(k, {})
        d[keys[-1]] = _coerce(value)
    return out


def _collect_yaml() -> dict:
    paths = [
        os.path.join(os.path.expanduser("~"), ".config", "mi355ml.yaml"),
        os.environ.get("MI355ML_CONFIG", ""),
    ]
    out: dict = {}
    for p in paths:
        if p and os.path.isfile(p):
            with open(p) as f:
                loaded = yaml.safe_load(f) or {}
            out = _deep_merge(out, loaded)
    return out


_base: dict = {}


def refresh() -> None:
    """Rebuild the config from defaults + yaml + env (reference: dask/config.py:402)."""
    global _base
    cfg = copy.deepcopy(DEFAULTS)
    cfg = _deep_merge(cfg, _collect_yaml())
    cfg = _deep_merge(cfg, _collect_env())
    _base = cfg


def _overlays() -> list:
    if not hasattr(_local, "overlays"):
        _local.overlays = []
    return _local.overlays


def get(key: str, default: Any = "__raise__") -> Any:
    """config.get("comm.backend") with dotted keys."""
    merged = _base
    for ov in _overlays():
        merged = _deep_merge(merged, ov)
    d: Any = merged
    for part in key.split("."):
        if not isinstance(d, dict) or part not in d:
            if default != "__raise__":
                return default
            raise KeyError(key)
        d = d[part]
    return d


@contextlib.contextmanager
def set(values: dict | None = None, **kwargs: Any) -> Iterator[None]:
    """Context manager overriding config keys (dotted or nested)."""
    upd: dict = {}
    items = dict(values or {})
    items.update({k.replace("__", "."): v for k, v in kwargs.items()})
    for key, v in items.items():
        d = upd
        parts = key.split(".")
        for p in parts[:-1]:
            d = d.setdefault(p, {})
        d[parts[-1]] = v
    _overlays().append(upd)
    try:
        yield
    finally:
        _overlays().remove(upd)


refresh()


def capture_overlays() -> list:
    """Snapshot this thread's config.set overlays (for propagating the
    active config context into worker threads — overlays are
    thread-local)."""
    return [copy.deepcopy(o) for o in _overlays()]


def install_overlays(overlays: list) -> None:
    """Install a snapshot from capture_overlays() in the current thread."""
    _local.overlays = list(overlays)
