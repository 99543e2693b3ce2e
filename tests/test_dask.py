"""Dask-on-Ray scheduler tests (reference: util/dask) — dask graphs
are plain dicts, so the scheduler is covered without dask installed."""
import operator
import time

import pytest

import ray_amd as ray  # noqa: F401
from ray_amd.util.dask import ray_dask_get


def inc(x):
    return x + 1


def add(x, y):
    return x + y


def test_ray_dask_get_basic(ray_start_regular):
    dsk = {
        "a": 1,
        "b": (inc, "a"),
        "c": (add, "b", 10),
        "d": (add, (inc, "c"), "b"),  # nested task
    }
    assert ray_dask_get(dsk, "d") == 13 + 2
    assert ray_dask_get(dsk, ["c", ["a", "b"]]) == [12, [1, 2]]


def test_ray_dask_get_parallel_fanout(ray_start_regular):
    def slow(x):
        time.sleep(0.4)
        return x * 2

    dsk = {"in": 5}
    for i in range(8):
        dsk[f"m{i}"] = (slow, "in")
    dsk["out"] = (sum, [(operator.add, f"m{i}", 0) for i in range(8)])
    t0 = time.time()
    assert ray_dask_get(dsk, "out") == 8 * 10
    # 8 x 0.4s of work finishing well under serial time proves fanout
    assert time.time() - t0 < 2.4


def test_ray_dask_get_cycle_detected(ray_start_regular):
    dsk = {"a": (inc, "b"), "b": (inc, "a")}
    with pytest.raises(ValueError, match="cycle"):
        ray_dask_get(dsk, "a")
