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
    t0 = time.time()
    refs = [a.work.remote(0.2) for _ in range(5)]
    assert ray.get(refs) == [0.2] * 5
    # concurrent: should take ~0.2s, not 1.0s
    assert time.time() - t0 < 0.9


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
