"""joblib integration (the reference's joblib/_dask.py bridge analog).

The reference registers a DaskDistributedBackend so sklearn's internal
`Parallel` loops fan out over the cluster (SURVEY §2.1 "Joblib bridge").
In the MI355X SPMD runtime the data-parallel axis is already consumed by
the 8 ranks — every mi355ml estimator is internally distributed — so the
bridge's remaining job is intra-rank task parallelism for CPU-side sklearn
work (e.g. ParallelPostFit over many small models). `register()` installs a
threading-based backend under the name "mi355ml" and returns it:

    import joblib, mi355ml.joblib_backend
    mi355ml.joblib_backend.register()
    with joblib.parallel_backend("mi355ml"):
        ...sklearn code...

Two backends are registered:

- "mi355ml": a threading backend for intra-rank CPU parallelism.
- "mi355ml_ranks": CROSS-RANK fan-out (the DaskDistributedBackend analog,
  joblib/_dask.py:143). SPMD contract: every rank must execute the same
  `Parallel(...)` call (the usual situation — sklearn code inside an SPMD
  program). Tasks are owned round-robin by rank; the owner executes its
  tasks eagerly during dispatch, every other rank stores a placeholder,
  and results are exchanged by one broadcast per task at retrieval time
  (joblib retrieves in dispatch order, so the broadcast schedule matches
  on all ranks). Tasks must not themselves issue collectives.
"""

from __future__ import annotations


class _RankResult:
    """Future-like: owner computed eagerly; .get() broadcasts the result."""

    def __init__(self, backend, owner: int, func):
        self._backend = backend
        self._owner = owner
        self._payload = None
        if backend._comm.rank == owner:
            try:
                self._payload = ("ok", func())
            except BaseException as e:  # rebroadcast as the task's outcome
                self._payload = ("err", e)

    def get(self, timeout=None):
        kind, val = self._backend._comm.broadcast_object(
            self._payload, src=self._owner)
        if kind == "err":
            raise val
        return val


def register():
    from joblib import register_parallel_backend
    from joblib._parallel_backends import (ParallelBackendBase,
                                           ThreadingBackend)

    class MI355MLBackend(ThreadingBackend):
        """Threading backend; GPU work inside tasks shares the rank's
        device/stream, CPU sklearn work gets real parallelism wherever the
        GIL is released (BLAS/C loops)."""

    class MI355MLRankBackend(ParallelBackendBase):
        """SPMD cross-rank fan-out (see module docstring)."""

        supports_retrieve_callback = False
        supports_timeout = False
        uses_threads = False
        supports_sharedmem = False

        def configure(self, n_jobs=1, parallel=None, **kwargs):
            from .parallel import comm as comm_mod
            self._comm = comm_mod.default()
            self._counter = 0
            self.parallel = parallel
            return self.effective_n_jobs(n_jobs)

        def effective_n_jobs(self, n_jobs):
            return max(1, self._comm.world_size)

        def submit(self, func, callback=None):
            owner = self._counter % self._comm.world_size
            self._counter += 1
            res = _RankResult(self, owner, func)
            if callback is not None:
                callback(res)
            return res

        def apply_async(self, func, callback=None):  # pragma: no cover
            return self.submit(func, callback)

        def retrieve_result(self, out, timeout=None):
            return out.get()

        def abort_everything(self, ensure_ready=True):
            if ensure_ready:
                self.configure(n_jobs=1, parallel=self.parallel)

    register_parallel_backend("mi355ml", MI355MLBackend)
    register_parallel_backend("mi355ml_ranks", MI355MLRankBackend)
    return MI355MLBackend
