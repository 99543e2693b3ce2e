"""Scalability-envelope checks (reference:
release/benchmarks/single_node + BASELINE.md scalability table).
Sizes scaled to CI time; the full-size numbers are measured by
tools/scalability_bench.py."""
import time

import numpy as np
import pytest

import ray_amd as ray


def test_many_args_to_one_task(ray_start_regular):
    # reference envelope: 10,000 object args (11.46 s); CI: 2,000
    refs = [ray.put(i) for i in range(2000)]

    @ray.remote
    def consume(lst):
        vals = ray.get(lst)
        return sum(vals)

    t0 = time.time()
    assert ray.get(consume.remote(refs), timeout=120) == sum(range(2000))
    assert time.time() - t0 < 60


def test_many_returns_from_one_task(ray_start_regular):
    # reference envelope: 3,000 returns (3.68 s); CI: 1,000
    n = 1000

    @ray.remote(num_returns=n)
    def produce():
        return tuple(range(n))

    refs = produce.remote()
    t0 = time.time()
    vals = ray.get(refs, timeout=120)
    assert vals == list(range(n))
    assert time.time() - t0 < 60


def test_wait_many_refs(ray_start_regular):
    # reference: ray.wait over 1k refs at 5.2 rounds/s
    @ray.remote
    def quick(i):
        return i

    refs = [quick.remote(i) for i in range(1000)]
    t0 = time.time()
    ready, not_ready = ray.wait(refs, num_returns=1000, timeout=120)
    assert len(ready) == 1000 and not not_ready
    assert time.time() - t0 < 60


def test_get_object_containing_many_refs(ray_start_regular):
    # reference: object containing 10k refs at 11.6/s; CI: 3k
    refs = [ray.put(i) for i in range(3000)]
    container = ray.put(refs)
    t0 = time.time()
    out = ray.get(container, timeout=120)
    assert len(out) == 3000
    assert ray.get(out[1234]) == 1234
    assert time.time() - t0 < 60


def test_many_queued_tasks(ray_start_regular):
    # reference envelope: 1M queued on one node; CI: 5,000 through 4 cpus
    @ray.remote
    def unit():
        return 1

    t0 = time.time()
    refs = [unit.remote() for _ in range(5000)]
    total = sum(ray.get(refs, timeout=300))
    dt = time.time() - t0
    assert total == 5000
    print(f"5000 queued tasks drained in {dt:.1f}s ({5000 / dt:.0f}/s)")


def test_large_object_roundtrip(ray_start_regular):
    # reference: 100 GiB max object; CI: 1 GiB zero-copy
    a = np.zeros(1 << 30, dtype=np.uint8)
    a[::65536] = 7
    t0 = time.time()
    ref = ray.put(a)
    b = ray.get(ref)
    dt = time.time() - t0
    assert b[65536] == 7 and b.nbytes == 1 << 30
    print(f"1 GiB put+get in {dt:.2f}s")
    assert dt < 60
