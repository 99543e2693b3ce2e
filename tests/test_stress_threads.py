"""Threaded stress (reference: python/ray/tests/test_threaded_actor.py,
test_actor_bounded_threads.py): the race-detection strategy for Python
paths is stress + invariants (docs/SANITIZERS.md)."""
import threading
import time

import numpy as np
import pytest

import ray_amd as ray


def test_threaded_actor_max_concurrency(ray_start_regular):
    @ray.remote(max_concurrency=8)
    class Conc:
        def __init__(self):
            self.lock = threading.Lock()
            self.active = 0
            self.peak = 0

        def work(self, ms):
            with self.lock:
                self.active += 1
                self.peak = max(self.peak, self.active)
            time.sleep(ms / 1000.0)
            with self.lock:
                self.active -= 1
            return True

        def peak_seen(self):
            return self.peak

    a = Conc.remote()
    refs = [a.work.remote(100) for _ in range(24)]
    assert all(ray.get(refs, timeout=120))
    # concurrency actually happened and stayed bounded
    peak = ray.get(a.peak_seen.remote())
    assert 2 <= peak <= 8, peak


def test_cross_thread_driver_calls(ray_start_regular):
    """ray.get/put from many driver threads concurrently."""
    @ray.remote
    def echo(x):
        return x

    errs = []

    def worker(tid):
        try:
            for i in range(20):
                v = {"tid": tid, "i": i, "a": np.arange(100)}
                ref = ray.put(v)
                got = ray.get(ref)
                assert got["tid"] == tid and got["i"] == i
                assert ray.get(echo.remote(i)) == i
        except BaseException as e:  # noqa
            errs.append((tid, repr(e)))

    ts = [threading.Thread(target=worker, args=(t,)) for t in range(8)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(120)
    assert not errs, errs


def test_concurrent_streaming_consumers(ray_start_regular):
    @ray.remote
    def gen(n):
        for i in range(n):
            yield i

    gens = [gen.options(num_returns="streaming").remote(20)
            for _ in range(4)]
    out = {}

    def consume(idx, g):
        out[idx] = [ray.get(r) for r in g]

    ts = [threading.Thread(target=consume, args=(i, g))
          for i, g in enumerate(gens)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(120)
    for i in range(4):
        assert out.get(i) == list(range(20)), out.get(i)
