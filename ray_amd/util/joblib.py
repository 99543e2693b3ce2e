"""joblib backend (reference: python/ray/util/joblib/) — scikit-learn
`n_jobs` parallelism over ray_amd tasks.

    from ray_amd.util.joblib import register_ray
    register_ray()
    with joblib.parallel_backend("ray_amd"):
        GridSearchCV(...).fit(X, y)
"""
from __future__ import annotations

from joblib._parallel_backends import ParallelBackendBase
from joblib.parallel import register_parallel_backend


class RayBackend(ParallelBackendBase):
    """Each joblib batch becomes one ray_amd task."""

    supports_timeout = True
    uses_threads = False
    supports_sharedmem = False

    def configure(self, n_jobs=1, parallel=None, **kwargs):
        import ray_amd as ray

        if not ray.is_initialized():
            ray.init(ignore_reinit_error=True)
        self.parallel = parallel
        return self.effective_n_jobs(n_jobs)

    def effective_n_jobs(self, n_jobs):
        import ray_amd as ray

        if not ray.is_initialized():
            ray.init(ignore_reinit_error=True)
        cpus = int(ray.cluster_resources().get("CPU", 1))
        if n_jobs is None:
            return 1
        if n_jobs < 0:
            return cpus
        return max(1, min(n_jobs, cpus))

    def apply_async(self, func, callback=None):
        import ray_amd as ray

        @ray.remote
        def _run_batch(payload):
            return payload()

        ref = _run_batch.remote(func)

        class _AsyncResult:
            def get(self, timeout=None):
                try:
                    v = ray.get(ref, timeout=timeout)
                except ray.exceptions.RayTaskError as e:
                    raise e.cause if e.cause is not None else e
                if callback is not None:
                    callback(v)
                return v

        return _AsyncResult()

    def abort_everything(self, ensure_ready=True):
        pass


def register_ray():
    register_parallel_backend("ray_amd", RayBackend)
    register_parallel_backend("ray", RayBackend)  # drop-in alias
