"""multiprocessing.Pool clone over ray_amd tasks (reference:
python/ray/util/multiprocessing/pool.py)."""
from __future__ import annotations

import itertools
from typing import Any, Callable, Iterable, List, Optional


class AsyncResult:
    def __init__(self, refs, single: bool):
        self._refs = refs
        self._single = single

    def get(self, timeout: Optional[float] = None):
        import ray_amd as ray

        vals = ray.get(self._refs, timeout=timeout)
        return vals[0] if self._single else vals

    def wait(self, timeout: Optional[float] = None):
        import ray_amd as ray

        ray.wait(self._refs, num_returns=len(self._refs), timeout=timeout)

    def ready(self) -> bool:
        import ray_amd as ray

        ready, _ = ray.wait(
            self._refs, num_returns=len(self._refs), timeout=0
        )
        return len(ready) == len(self._refs)

    def successful(self) -> bool:
        try:
            self.get(timeout=0.001)
            return True
        except Exception:
            return False


class Pool:
    def __init__(self, processes: Optional[int] = None, initializer=None,
                 initargs=(), ray_address=None):
        import ray_amd as ray

        self._ray = ray
        if not ray.is_initialized():
            ray.init(address=ray_address)
        self._processes = processes or int(ray.cluster_resources().get("CPU", 4))
        self._initializer = initializer
        self._initargs = initargs
        self._closed = False

    def _task(self, func):
        ray = self._ray
        initializer = self._initializer
        initargs = self._initargs

        @ray.remote
        def _run(batch):
            import builtins

            if initializer is not None and not getattr(
                builtins, "_ray_amd_pool_init", False
            ):
                initializer(*initargs)
                builtins._ray_amd_pool_init = True
            return [func(*a) if isinstance(a, tuple) else func(a) for a in batch]

        return _run

    def apply(self, func: Callable, args=(), kwds=None):
        return self.apply_async(func, args, kwds).get()

    def apply_async(self, func, args=(), kwds=None, callback=None,
                    error_callback=None):
        ray = self._ray

        @ray.remote
        def _one(a, k):
            return func(*a, **(k or {}))

        return AsyncResult([_one.remote(args, kwds)], single=True)

    def map(self, func: Callable, iterable: Iterable, chunksize=None) -> List[Any]:
        return self.map_async(func, iterable, chunksize).get()

    def map_async(self, func, iterable, chunksize=None):
        items = list(iterable)
        cs = chunksize or max(1, len(items) // (self._processes * 4) or 1)
        run = self._task(func)
        refs = [
            run.remote(items[i : i + cs]) for i in range(0, len(items), cs)
        ]
        return _FlattenResult(refs)

    def starmap(self, func, iterable, chunksize=None):
        items = [tuple(x) for x in iterable]
        return self.map(func, items, chunksize)

    def imap(self, func, iterable, chunksize=1):
        run = self._task(func)
        items = list(iterable)
        refs = [run.remote(items[i : i + chunksize])
                for i in range(0, len(items), chunksize)]
        for r in refs:
            for v in self._ray.get(r):
                yield v

    def imap_unordered(self, func, iterable, chunksize=1):
        run = self._task(func)
        items = list(iterable)
        refs = [run.remote(items[i : i + chunksize])
                for i in range(0, len(items), chunksize)]
        remaining = list(refs)
        while remaining:
            ready, remaining = self._ray.wait(remaining, num_returns=1)
            for v in self._ray.get(ready[0]):
                yield v

    def close(self):
        self._closed = True

    def terminate(self):
        self._closed = True

    def join(self):
        pass

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


class _FlattenResult(AsyncResult):
    def __init__(self, refs):
        super().__init__(refs, single=False)

    def get(self, timeout=None):
        batches = super().get(timeout)
        return list(itertools.chain.from_iterable(batches))
