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

Process-level fan-out (one backend worker per GPU rank) is intentionally
NOT provided: nested torch.distributed collectives inside joblib tasks
would deadlock; use the model_selection searches for cross-rank trials.
"""

from __future__ import annotations


def register():
    from joblib import register_parallel_backend
    from joblib._parallel_backends import ThreadingBackend

    class MI355MLBackend(ThreadingBackend):
        """Threading backend; GPU work inside tasks shares the rank's
        device/stream, CPU sklearn work gets real parallelism wherever the
        GIL is released (BLAS/C loops)."""

    register_parallel_backend("mi355ml", MI355MLBackend)
    return MI355MLBackend
