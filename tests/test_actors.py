"""Actor tests (reference model: python/ray/tests/test_actor*.py)."""
import time

import pytest

import ray_amd as ray


@ray.remote
class Counter:
    def __init__(self, start=0):
        self.v = start

    def incr(self, k=1):
        self.v += k
        return self.v

    def get(self):
        return self.v

    def fail(self):
        raise RuntimeError("actor method failed")


def test_actor_basic(ray_start_regular):
    c = Counter.remote(5)
    assert ray.get(c.incr.remote()) == 6
    assert ray.get(c.incr.remote(4)) == 10
    assert ray.get(c.get.remote()) == 10


def test_actor_ordering(ray_start_regular):
    c = Counter.remote()
    refs = [c.incr.remote() for _ in range(200)]
    assert ray.get(refs) == list(range(1, 201))


def test_actor_method_exception(ray_start_regular):
    c = Counter.remote()
    with pytest.raises(ray.exceptions.RayTaskError, match="actor method failed"):
        ray.get(c.fail.remote())
    # actor still alive
    assert ray.get(c.incr.remote()) == 1


def test_actor_init_args_with_refs(ray_start_regular):
    ref = ray.put(100)
    c = Counter.remote(ref)
    assert ray.get(c.get.remote()) == 100


def test_named_actor(ray_start_regular):
    Counter.options(name="named_c").remote(7)
    h = ray.get_actor("named_c")
    assert ray.get(h.get.remote()) == 7


def test_get_actor_missing(ray_start_regular):
    with pytest.raises(ValueError):
        ray.get_actor("does_not_exist")


def test_named_actor_duplicate(ray_start_regular):
    Counter.options(name="dup").remote()
    time.sleep(0.2)
    with pytest.raises(Exception):
        h2 = Counter.options(name="dup").remote()
        ray.get(h2.get.remote())


def test_get_if_exists(ray_start_regular):
    a = Counter.options(name="gie", get_if_exists=True).remote(3)
    ray.get(a.get.remote())
    b = Counter.options(name="gie", get_if_exists=True).remote(99)
    assert ray.get(b.get.remote()) == 3  # existing actor reused


def test_kill_actor(ray_start_regular):
    c = Counter.remote()
    assert ray.get(c.incr.remote()) == 1
    ray.kill(c)
    with pytest.raises(ray.exceptions.RayActorError):
        ray.get(c.incr.remote(), timeout=20)


def test_actor_handle_passing(ray_start_regular):
    c = Counter.remote()

    @ray.remote
    def use(handle):
        return ray.get(handle.incr.remote(10))

    assert ray.get(use.remote(c)) == 10
    assert ray.get(c.get.remote()) == 10


def test_actor_restart(ray_start_regular):
    @ray.remote(max_restarts=1)
    class Flaky:
        def __init__(self):
            self.v = 0

        def die(self):
            import os

            os._exit(1)

        def ping(self):
            self.v += 1
            return self.v

    f = Flaky.remote()
    assert ray.get(f.ping.remote()) == 1
    try:
        ray.get(f.die.remote(), timeout=20)
    except Exception:
        pass
    # should be restarted with fresh state
    deadline = time.time() + 30
    while time.time() < deadline:
        try:
            assert ray.get(f.ping.remote(), timeout=10) == 1
            break
        except ray.exceptions.RayError:
            # in-flight calls racing the restart may surface as actor
            # or transient system errors; keep probing until the fresh
            # instance answers
            time.sleep(0.2)
    else:
        pytest.fail("actor did not restart")


def test_async_actor(ray_start_regular):
    @ray.remote
    class AsyncActor:
        async def work(self, t):
            import asyncio

            await asyncio.sleep(t)
            return t

    a = AsyncActor.remote()
    ray.get(a.work.remote(0.01))  # actor up before timing
    t0 = time.time()
    refs = [a.work.remote(0.3) for _ in range(5)]
    assert ray.get(refs) == [0.3] * 5
    # concurrent: ~0.3s, serial would be 1.5s (margin for loaded CI)
    assert time.time() - t0 < 1.2


def test_max_concurrency_threaded(ray_start_regular):
    @ray.remote(max_concurrency=4)
    class Slow:
        def work(self):
            time.sleep(0.2)
            return 1

    s = Slow.remote()
    t0 = time.time()
    ray.get([s.work.remote() for _ in range(4)])
    assert time.time() - t0 < 0.7


def test_exit_actor(ray_start_regular):
    @ray.remote
    class Quitter:
        def quit(self):
            ray.exit_actor()

        def ping(self):
            return "pong"

    q = Quitter.remote()
    assert ray.get(q.ping.remote()) == "pong"
    ray.get(q.quit.remote())
    time.sleep(0.5)
    with pytest.raises(ray.exceptions.RayActorError):
        ray.get(q.ping.remote(), timeout=20)


def test_actor_creation_error_surfaces(ray_start_regular):
    @ray.remote
    class Bad:
        def __init__(self):
            raise RuntimeError("init failed")

        def ping(self):
            return 1

    b = Bad.remote()
    with pytest.raises(ray.exceptions.RayActorError):
        ray.get(b.ping.remote(), timeout=30)


def test_fractional_gpu_device_assignment():
    """Advisor (round 1): num_gpus=0.5 used to floor to 0 device ids —
    no HIP_VISIBLE_DEVICES isolation. Fractional actors must share a
    single device id and accounting must queue (not under-assign) when
    no device has the fraction free. GPUs are faked (no HIP needed)."""
    import ray_amd as ray

    ray.init(num_cpus=8, num_gpus=2, ignore_reinit_error=True)
    try:
        @ray.remote(num_gpus=0.5)
        class Half:
            def ids(self):
                return ray.get_gpu_ids()

        actors = [Half.remote() for _ in range(4)]
        ids = ray.get([a.ids.remote() for a in actors], timeout=60)
        # each fractional actor sees exactly one device
        assert all(len(x) == 1 for x in ids), ids
        # 4 x 0.5 packs onto 2 devices, 2 actors per device
        from collections import Counter as C

        counts = C(x[0] for x in ids)
        assert sorted(counts.values()) == [2, 2], ids

        # a whole-GPU actor cannot be placed now: 0 fully-free devices
        @ray.remote(num_gpus=1)
        class Whole:
            def ids(self):
                return ray.get_gpu_ids()

        w = Whole.remote()
        import time as _t

        ready, _ = ray.wait([w.ids.remote()], timeout=2)
        assert not ready  # queued, not silently under-assigned
        # free two halves on one device -> the whole actor still cannot
        # fit (each device has 0.5 used at best after killing 2 on the
        # same device frees 1.0 on it)
        victim_dev = ids[0][0]
        for a, x in zip(actors, ids):
            if x[0] == victim_dev:
                ray.kill(a)
        _t.sleep(1.0)
        got = ray.get(w.ids.remote(), timeout=60)
        assert got == [victim_dev]
    finally:
        ray.shutdown()


def test_ray_method_num_returns(ray_start_regular):
    """Advisor (round 1): @ray.method(num_returns=N) was a silent no-op."""

    @ray.remote
    class Pair:
        @ray.method(num_returns=2)
        def two(self):
            return 1, 2

        def one(self):
            return (3, 4)

    p = Pair.remote()
    a, b = p.two.remote()
    assert ray.get(a) == 1 and ray.get(b) == 2
    # call-site options still override the method default
    ref = p.two.options(num_returns=1).remote()
    assert ray.get(ref) == (1, 2)
    # and the handle survives serialization with its method options
    import cloudpickle

    p2 = cloudpickle.loads(cloudpickle.dumps(p))
    c, d = p2.two.remote()
    assert ray.get([c, d]) == [1, 2]


def test_actor_max_task_retries_resubmits(ray_start_regular):
    """max_task_retries > 0 (reference: actor_task_submitter.cc:597):
    a call in flight when the actor dies is resubmitted to the
    restarted instance instead of failing with ActorUnavailableError."""
    import os as _os

    @ray.remote(max_restarts=2, max_task_retries=2)
    class Flaky:
        def __init__(self):
            self.n = 0

        def die_then_answer(self, marker):
            # first instance dies mid-call; the restarted one answers
            if not _os.path.exists(marker):
                open(marker, "w").write("x")
                _os._exit(1)
            return "answered"

    import tempfile

    marker = tempfile.mktemp()
    a = Flaky.remote()
    assert ray.get(a.die_then_answer.remote(marker), timeout=120) == "answered"

    # default (max_task_retries=0) stays at-most-once
    @ray.remote(max_restarts=2)
    class Flaky0:
        def die(self, marker):
            if not _os.path.exists(marker):
                open(marker, "w").write("x")
                _os._exit(1)
            return "no"

    marker2 = tempfile.mktemp()
    b = Flaky0.remote()
    with pytest.raises(ray.exceptions.RayError):
        ray.get(b.die.remote(marker2), timeout=60)


def test_profiler_runtime_env_wrapper(ray_start_regular):
    """Profiler runtime-env plugin (reference:
    runtime_env/rocprof_sys.py:17 — the worker launches UNDER the
    profiler wrapper). Verified with the generic wrapper (`env VAR=x`)
    so the test runs without a GPU; {'rocprof': {...}} expands to a
    rocprofv3 wrapper the same way."""
    import os as _os

    @ray.remote(runtime_env={"_wrapper_cmd": ["env", "RAY_AMD_WRAPPED=yes"]})
    class Probed:
        def wrapped(self):
            return _os.environ.get("RAY_AMD_WRAPPED")

    a = Probed.remote()
    assert ray.get(a.wrapped.remote(), timeout=60) == "yes"

    # the rocprof sugar produces a rocprofv3 command line
    from ray_amd._core.runtime import _profiler_cmd

    cmd = _profiler_cmd({"rocprof": {"output_dir": "/tmp/p",
                                     "args": ["--kernel-trace"]}})
    assert cmd == ["rocprofv3", "--kernel-trace", "-d", "/tmp/p", "--"]
