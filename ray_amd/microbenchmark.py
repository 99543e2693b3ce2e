"""Core microbenchmarks (reference: python/ray/_private/ray_perf.py:95,
CLI `ray microbenchmark`, scripts.py:2275). Same metric names as
release/perf_metrics/microbenchmark.json so results compare 1:1 with
BASELINE.md."""
from __future__ import annotations

import time
from typing import Callable, List, Tuple

import numpy as np


def timeit(name: str, fn: Callable, multiplier: int = 1,
           duration: float = 2.0) -> Tuple[str, float]:
    # warmup
    fn()
    count = 0
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < duration:
        fn()
        count += 1
    dt = time.perf_counter() - t0
    rate = count * multiplier / dt
    print(f"{name} per second: {rate:.2f}")
    return (name, rate)


def main(duration: float = 2.0) -> List[Tuple[str, float]]:
    import ray_amd as ray

    results = []
    ray.init(ignore_reinit_error=True)

    arr = np.zeros(100 * 1024 * 1024, dtype=np.uint8)  # 100 MB

    # Honest shm round-trip (the reference's "get calls" is a plasma
    # round-trip per call): force the put through the shm store and
    # reset the memory-store entry between gets so every measured get
    # re-maps + re-deserializes — NOT a dict hit on the cached value.
    from ray_amd._core import runtime as _rtmod

    rt = _rtmod.global_runtime()
    value = rt.put(0, _force_store=True)
    _store_entry = rt.memory_store[value.id]

    def get_roundtrip():
        for _ in range(100):
            ray.get(value)
            rt.memory_store[value.id] = _store_entry
            rt._mmaps.pop(value.id, None)

    results.append(timeit("single client get calls", get_roundtrip, 100, duration))
    results.append(timeit("single client put calls", lambda: [ray.put(0) for _ in range(100)], 100, duration))
    results.append(
        timeit("single client put gigabytes",
               lambda: ray.put(arr), 0.1, duration)
    )

    @ray.remote
    def small_task():
        return b"ok"

    results.append(
        timeit("single client tasks sync",
               lambda: ray.get(small_task.remote()), 1, duration)
    )
    results.append(
        timeit(
            "single client tasks async",
            lambda: ray.get([small_task.remote() for _ in range(1000)]),
            1000,
            duration,
        )
    )

    @ray.remote
    class Client:
        def __init__(self):
            pass

        def run_tasks(self, n):
            ray.get([small_task.remote() for _ in range(n)])

        def run_puts(self, n):
            import numpy as _np

            for _ in range(n):
                ray.put(_np.zeros(10 * 1024 * 1024, dtype=_np.uint8))

    clients = [Client.remote() for _ in range(4)]
    ray.get([c.run_tasks.remote(10) for c in clients])  # warm pools
    results.append(
        timeit(
            "multi client tasks async",
            lambda: ray.get([c.run_tasks.remote(250) for c in clients]),
            1000,
            duration,
        )
    )
    results.append(
        timeit(
            "multi client put gigabytes",
            lambda: ray.get([c.run_puts.remote(1) for c in clients]),
            4 * 10 * 1024 * 1024 / 1e9,
            duration,
        )
    )

    @ray.remote
    class Actor:
        def small_value(self):
            return b"ok"

    a = Actor.remote()
    results.append(
        timeit("1:1 actor calls sync",
               lambda: ray.get(a.small_value.remote()), 1, duration)
    )
    results.append(
        timeit(
            "1:1 actor calls async",
            lambda: ray.get([a.small_value.remote() for _ in range(1000)]),
            1000,
            duration,
        )
    )
    ac = Actor.options(max_concurrency=16).remote()
    results.append(
        timeit(
            "1:1 actor calls concurrent",
            lambda: ray.get([ac.small_value.remote() for _ in range(1000)]),
            1000,
            duration,
        )
    )
    actors = [Actor.remote() for _ in range(8)]
    results.append(
        timeit(
            "1:n actor calls async",
            lambda: ray.get(
                [b.small_value.remote() for b in actors for _ in range(125)]
            ),
            1000,
            duration,
        )
    )

    @ray.remote
    class Caller:
        def __init__(self, targets):
            self.targets = targets

        def run(self, n):
            ray.get(
                [t.small_value.remote() for t in self.targets for _ in range(n)]
            )

    callers = [Caller.remote(actors) for _ in range(4)]
    ray.get([c.run.remote(2) for c in callers])
    results.append(
        timeit(
            "n:n actor calls async",
            lambda: ray.get([c.run.remote(31) for c in callers]),
            31 * 8 * 4,
            duration,
        )
    )

    @ray.remote
    class AsyncActor:
        async def small_value(self):
            return b"ok"

    aa = AsyncActor.remote()
    results.append(
        timeit("1:1 async-actor calls sync",
               lambda: ray.get(aa.small_value.remote()), 1, duration)
    )
    results.append(
        timeit(
            "1:1 async-actor calls async",
            lambda: ray.get([aa.small_value.remote() for _ in range(1000)]),
            1000,
            duration,
        )
    )

    from ray_amd.util import placement_group, remove_placement_group

    def pg_cycle():
        pg = placement_group([{"CPU": 0.001}])
        pg.wait(10)
        remove_placement_group(pg)

    results.append(timeit("placement group create/removal", pg_cycle, 1, duration))

    ray.shutdown()
    return results


if __name__ == "__main__":
    import sys

    d = float(sys.argv[1]) if len(sys.argv) > 1 else 2.0
    main(d)
