"""Core task/object API tests (reference model:
python/ray/tests/test_basic.py)."""
import time

import numpy as np
import pytest

import ray_amd as ray


def test_put_get(ray_start_regular):
    for v in [1, "x", None, [1, 2, {"a": (3, 4)}], b"bytes"]:
        assert ray.get(ray.put(v)) == v


def test_put_get_numpy_zero_copy(ray_start_regular):
    a = np.random.rand(512, 512)
    ref = ray.put(a)
    b = ray.get(ref)
    np.testing.assert_array_equal(a, b)
    assert not b.flags.writeable  # zero-copy view onto shm
    # repeated get returns cached value
    c = ray.get(ref)
    assert c is b


def test_simple_task(ray_start_regular):
    @ray.remote
    def f(x):
        return x * 2

    assert ray.get(f.remote(21)) == 42


def test_task_many(ray_start_regular):
    @ray.remote
    def f(i):
        return i

    assert ray.get([f.remote(i) for i in range(300)]) == list(range(300))


def test_task_args_kwargs_defaults(ray_start_regular):
    @ray.remote
    def f(a, b=10, *args, **kwargs):
        return a + b + sum(args) + kwargs.get("c", 0)

    assert ray.get(f.remote(1)) == 11
    assert ray.get(f.remote(1, 2)) == 3
    assert ray.get(f.remote(1, 2, 3, c=4)) == 10


def test_ref_as_arg_resolved(ray_start_regular):
    @ray.remote
    def plus1(x):
        return x + 1

    r = plus1.remote(plus1.remote(plus1.remote(0)))
    assert ray.get(r) == 3


def test_nested_refs_not_resolved(ray_start_regular):
    @ray.remote
    def check(lst):
        return all(isinstance(x, ray.ObjectRef) for x in lst)

    refs = [ray.put(i) for i in range(3)]
    assert ray.get(check.remote(refs))


def test_large_arg_and_return(ray_start_regular):
    @ray.remote
    def echo(a):
        return a * 2

    a = np.ones((1500, 1500), dtype=np.float32)
    out = ray.get(echo.remote(a))
    assert out.shape == (1500, 1500) and out[0, 0] == 2.0


def test_num_returns(ray_start_regular):
    @ray.remote(num_returns=3)
    def three():
        return 1, 2, 3

    a, b, c = three.remote()
    assert ray.get([a, b, c]) == [1, 2, 3]


def test_task_exception(ray_start_regular):
    @ray.remote
    def boom():
        raise ValueError("kaboom")

    with pytest.raises(ray.exceptions.RayTaskError, match="kaboom"):
        ray.get(boom.remote())


def test_exception_in_dependency_propagates(ray_start_regular):
    @ray.remote
    def boom():
        raise ValueError("kaboom")

    @ray.remote
    def consume(x):
        return x

    with pytest.raises(Exception, match="kaboom"):
        ray.get(consume.remote(boom.remote()))


def test_wait(ray_start_regular):
    @ray.remote
    def slow(t):
        time.sleep(t)
        return t

    refs = [slow.remote(0.05), slow.remote(10)]
    ready, not_ready = ray.wait(refs, num_returns=1, timeout=5)
    assert ready == [refs[0]] and not_ready == [refs[1]]

    ready, not_ready = ray.wait([slow.remote(0.01)], num_returns=1, timeout=5)
    assert len(ready) == 1 and not not_ready


def test_wait_timeout(ray_start_regular):
    @ray.remote
    def slow():
        time.sleep(30)

    ready, not_ready = ray.wait([slow.remote()], num_returns=1, timeout=0.2)
    assert not ready and len(not_ready) == 1


def test_get_timeout(ray_start_regular):
    @ray.remote
    def slow():
        time.sleep(30)

    with pytest.raises(ray.exceptions.GetTimeoutError):
        ray.get(slow.remote(), timeout=0.2)


def test_options_name(ray_start_regular):
    @ray.remote
    def f():
        return 1

    assert ray.get(f.options(name="custom").remote()) == 1


def test_runtime_env_env_vars(ray_start_regular):
    import os

    @ray.remote
    def read_env():
        return os.environ.get("MY_TEST_VAR")

    v = ray.get(
        read_env.options(runtime_env={"env_vars": {"MY_TEST_VAR": "42"}}).remote()
    )
    assert v == "42"


def test_cluster_resources(ray_start_regular):
    res = ray.cluster_resources()
    assert res["CPU"] == 4


def test_put_of_object_ref_fails(ray_start_regular):
    with pytest.raises(TypeError):
        ray.put(ray.put(1))


def test_async_task_function(ray_start_regular):
    @ray.remote
    async def afn(x):
        import asyncio

        await asyncio.sleep(0.01)
        return x + 1

    assert ray.get(afn.remote(1)) == 2


def test_state_api_and_cli(ray_start_regular):
    @ray.remote
    def f():
        return 1

    ray.get([f.remote() for _ in range(10)])

    @ray.remote
    class A:
        def ping(self):
            return 1

    a = A.remote()
    ray.get(a.ping.remote())
    import time as _t

    _t.sleep(0.3)
    from ray_amd.util import state as state_api

    nodes = state_api.list_nodes()
    assert nodes and nodes[0]["state"] == "ALIVE"
    actors = state_api.list_actors()
    assert any(x["class_name"] == "A" for x in actors)
    objs = state_api.list_objects()
    assert "num_objects_in_store" in objs[0]


def test_streaming_generator_task(ray_start_regular):
    @ray.remote
    def gen(n):
        for i in range(n):
            yield i * 10

    g = gen.options(num_returns="streaming").remote(5)
    vals = [ray.get(ref) for ref in g]
    assert vals == [0, 10, 20, 30, 40]


def test_streaming_generator_large_items(ray_start_regular):
    @ray.remote
    def gen():
        for i in range(3):
            yield np.full((600, 600), i, dtype=np.float64)  # ~2.9MB

    g = gen.options(num_returns="streaming").remote()
    arrs = [ray.get(r) for r in g]
    assert [int(a[0, 0]) for a in arrs] == [0, 1, 2]


def test_streaming_generator_error(ray_start_regular):
    @ray.remote
    def gen():
        yield 1
        raise ValueError("stream boom")

    g = gen.options(num_returns="streaming").remote()
    first = ray.get(next(g))
    assert first == 1
    with pytest.raises(Exception, match="stream boom"):
        for r in g:
            ray.get(r)


def test_streaming_actor_method(ray_start_regular):
    @ray.remote
    class Gen:
        def stream(self, n):
            for i in range(n):
                yield i

    g = Gen.remote()
    it = g.stream.options(num_returns="streaming").remote(4)
    assert [ray.get(r) for r in it] == [0, 1, 2, 3]


def test_streaming_ids_do_not_collide_with_object_ids(ray_start_regular):
    """Regression (advisor, round 1): stream-item oids were
    task_id(nonce8+low4(task_seq)) + idx4, byte-identical to the normal
    object id with obj_seq == task_seq, so a streaming item silently
    overwrote a live object's memory-store entry. Put refs while
    streaming tasks run through the same seq range; every put must
    still read back its own value."""

    @ray.remote
    def gen(n):
        for i in range(n):
            yield ("stream", i)

    held = [ray.put(("kept", i)) for i in range(25)]
    for _ in range(25):
        g = gen.options(num_returns="streaming").remote(3)
        out = [ray.get(r) for r in g]
        assert out == [("stream", 0), ("stream", 1), ("stream", 2)]
    for i, ref in enumerate(held):
        assert ray.get(ref) == ("kept", i)


def test_metrics_api(ray_start_regular):
    from ray_amd.util import metrics

    c = metrics.Counter("test_requests", "requests", ("route",))
    c.inc(1, {"route": "/a"})
    c.inc(2, {"route": "/a"})
    g = metrics.Gauge("test_inflight", "inflight")
    g.set(5)
    h = metrics.Histogram("test_lat", "latency", boundaries=[1, 10])
    h.observe(0.5)
    h.observe(5)
    h.observe(50)
    text = metrics.export_text()
    assert "test_requests" in text and "3.0" in text
    assert "test_inflight 5" in text
    assert "test_lat_count 3" in text


def test_multiprocessing_pool(ray_start_regular):
    from ray_amd.util.multiprocessing import Pool

    with Pool(processes=2) as pool:
        out = pool.map(lambda x: x * x, range(20))
        assert out == [x * x for x in range(20)]
        assert pool.apply(lambda a, b: a + b, (3, 4)) == 7
        assert sorted(pool.imap_unordered(lambda x: -x, range(5))) == [-4, -3, -2, -1, 0]


def test_internal_kv(ray_start_regular):
    from ray_amd.experimental import internal_kv as kv

    assert kv._internal_kv_initialized()
    kv._internal_kv_put(b"k1", b"v1")
    assert kv._internal_kv_get(b"k1") == b"v1"
    assert kv._internal_kv_exists(b"k1")
    assert b"k1" in kv._internal_kv_list(b"k")
    kv._internal_kv_del(b"k1")
    assert kv._internal_kv_get(b"k1") is None


def test_dashboard_api(ray_start_regular):
    import anyio
    import httpx

    from ray_amd.dashboard import build_asgi_app

    @ray.remote
    class DashActor:
        def ping(self):
            return 1

    a = DashActor.remote()
    ray.get(a.ping.remote())

    async def go():
        transport = httpx.ASGITransport(app=build_asgi_app())
        async with httpx.AsyncClient(transport=transport,
                                     base_url="http://d") as client:
            r = await client.get("/api/cluster_status")
            assert r.status_code == 200
            body = r.json()
            assert body["nodes"] >= 1
            assert body["actors_alive"] >= 1
            r = await client.get("/api/nodes")
            assert r.json()[0]["Alive"]
            r = await client.get("/")
            assert b"ray_amd" in r.content

    anyio.run(go)


def test_runtime_env_working_dir(ray_start_regular, tmp_path):
    import os

    mod_dir = tmp_path / "my_wd"
    mod_dir.mkdir()
    (mod_dir / "wd_module.py").write_text("MAGIC = 'from-working-dir'\n")
    (mod_dir / "data.txt").write_text("hello-wd")

    @ray.remote
    def use_wd():
        import wd_module

        return wd_module.MAGIC, open("data.txt").read()

    magic, data = ray.get(
        use_wd.options(
            runtime_env={"working_dir": str(mod_dir)}
        ).remote(),
        timeout=60,
    )
    assert magic == "from-working-dir"
    assert data == "hello-wd"


def test_runtime_env_py_modules(ray_start_regular, tmp_path):
    pkg = tmp_path / "mypkg"
    pkg.mkdir()
    (pkg / "extra_mod.py").write_text("VALUE = 42\n")

    @ray.remote
    class UsesModule:
        def get(self):
            import extra_mod

            return extra_mod.VALUE

    a = UsesModule.options(
        runtime_env={"py_modules": [str(pkg)]}
    ).remote()
    assert ray.get(a.get.remote(), timeout=60) == 42


def test_second_driver_connects(ray_start_regular):
    """A separate driver process connects to the running cluster by
    address and shares named actors (reference: multi-driver clusters)."""
    import subprocess
    import sys

    @ray.remote
    class Shared:
        def __init__(self):
            self.v = 41

        def bump(self):
            self.v += 1
            return self.v

    Shared.options(name="shared_counter", lifetime="detached").remote()
    rt = ray.api._rt.global_runtime()
    script = f"""
import ray_amd as ray
ray.init(address={rt.session_dir!r})
h = ray.get_actor("shared_counter")
print("RESULT", ray.get(h.bump.remote(), timeout=30))
ray.shutdown(_exiting_interpreter=True)
"""
    out = subprocess.run(
        [sys.executable, "-c", script], capture_output=True, text=True,
        timeout=120,
    )
    assert "RESULT 42" in out.stdout, out.stdout + out.stderr
    # first driver still sees the state
    h = ray.get_actor("shared_counter")
    assert ray.get(h.bump.remote()) == 43


def test_cancel_queued_task(ray_start_regular):
    @ray.remote
    def blocker():
        time.sleep(5)
        return "done"

    @ray.remote
    def victim():
        return "ran"

    # saturate the 4 CPUs, then queue a victim and cancel it
    blockers = [blocker.remote() for _ in range(4)]
    time.sleep(0.3)
    v = victim.remote()
    ray.cancel(v)
    with pytest.raises(ray.exceptions.TaskCancelledError):
        ray.get(v, timeout=30)
    assert ray.get(blockers, timeout=30) == ["done"] * 4


def test_joblib_backend(ray_start_regular):
    import joblib

    from ray_amd.util.joblib import register_ray

    register_ray()
    with joblib.parallel_backend("ray_amd", n_jobs=2):
        out = joblib.Parallel()(joblib.delayed(lambda x: x * x)(i)
                                for i in range(12))
    assert out == [i * i for i in range(12)]


def test_joblib_sklearn(ray_start_regular):
    import joblib
    import numpy as np
    from sklearn.ensemble import RandomForestClassifier

    from ray_amd.util.joblib import register_ray

    register_ray()
    X = np.random.rand(80, 5)
    y = (X[:, 0] > 0.5).astype(int)
    with joblib.parallel_backend("ray_amd", n_jobs=2):
        clf = RandomForestClassifier(n_estimators=8, n_jobs=2).fit(X, y)
    assert clf.score(X, y) > 0.8


def test_state_get_log(ray_start_regular):
    @ray.remote
    def noisy():
        print("hello-from-worker-log")
        return 1

    ray.get([noisy.remote() for _ in range(3)])
    import time as _t

    _t.sleep(0.3)
    from ray_amd.util import state as state_api

    logs = state_api.list_logs()["worker_out"]
    assert any(f.startswith("worker_") for f in logs)
    found = False
    for f in logs:
        if f.startswith("worker_"):
            for line in state_api.get_log(filename=f):
                if "hello-from-worker-log" in line:
                    found = True
    assert found


def test_retry_exceptions(ray_start_regular, tmp_path):
    marker = str(tmp_path / "attempts")

    @ray.remote(max_retries=3, retry_exceptions=True)
    def flaky(marker):
        import os

        n = int(open(marker).read()) if os.path.exists(marker) else 0
        open(marker, "w").write(str(n + 1))
        if n < 2:
            raise RuntimeError(f"attempt {n} fails")
        return n

    assert ray.get(flaky.remote(marker), timeout=60) == 2


def test_no_retry_exceptions_by_default(ray_start_regular):
    @ray.remote
    def boom():
        raise RuntimeError("once")

    with pytest.raises(ray.exceptions.RayTaskError):
        ray.get(boom.remote(), timeout=30)


def test_pubsub_bus(ray_start_regular):
    """GCS pub/sub: cross-process fan-out, no replay for late subs."""
    import queue

    from ray_amd.util import pubsub

    with pubsub.Subscriber("events") as sub:
        # another process (a task) publishes
        @ray.remote
        def pub(msg):
            from ray_amd.util import pubsub as ps

            return ps.publish("events", msg)

        reached = ray.get(pub.remote({"k": 1}), timeout=30)
        assert reached >= 1
        assert sub.poll(timeout=10) == {"k": 1}

        # driver-side publish also delivers
        pubsub.publish("events", "plain-string")
        assert sub.poll(timeout=10) == "plain-string"

    # closed subscriber no longer receives
    assert pubsub.publish("events", "late") == 0

    with pubsub.Subscriber("events") as sub2:
        with pytest.raises(queue.Empty):
            sub2.poll(timeout=0.2)  # no replay of earlier messages


def test_object_locations_and_dynamic_resources(ray_start_regular):
    from ray_amd import experimental as exp

    # locations: a big object lives in this node's store
    big = ray.put(np.zeros(500_000, dtype=np.uint8))
    loc = exp.get_object_locations([big])[big]
    assert loc["object_size"] and loc["object_size"] >= 500_000
    assert len(loc["node_ids"]) == 1

    # dynamic resource: create at runtime, schedule on it, delete
    exp.set_resource("tokens", 2)
    time.sleep(0.3)
    assert ray.cluster_resources().get("tokens") == 2

    @ray.remote(resources={"tokens": 1})
    def use():
        return "ok"

    assert ray.get(use.remote(), timeout=30) == "ok"
    exp.set_resource("tokens", 0)
    time.sleep(0.3)
    assert "tokens" not in ray.cluster_resources()


def test_ray_config_flags(monkeypatch):
    """Central flag table (reference: ray_config_def.h RayConfig)."""
    from ray_amd._config import RayConfig, config

    desc = RayConfig.describe()
    assert len(desc) >= 10
    assert all("env" in v and "doc" in v for v in desc.values())
    assert config.lease_request_cap == 16
    monkeypatch.setenv("RAY_AMD_LEASE_REQUEST_CAP", "4")
    config.reload()
    assert config.lease_request_cap == 4
    monkeypatch.delenv("RAY_AMD_LEASE_REQUEST_CAP")
    config.reload()
    with pytest.raises(AttributeError):
        config.no_such_flag


def test_streaming_generator_backpressure(ray_start_regular, tmp_path):
    """A fast producer pauses once the default cap (64) of yielded
    items sit unconsumed (reference:
    generator_backpressure_num_objects). Workers read the cap from
    their own env at spawn, so the test rides the default."""
    marker = str(tmp_path / "produced")
    n = 200

    @ray.remote(num_returns="streaming")
    def produce(marker=marker, n=n):
        for i in range(n):
            with open(marker, "w") as f:
                f.write(str(i + 1))
            yield i

    gen = produce.remote()
    it = iter(gen)
    first = ray.get(next(it), timeout=30)
    assert first == 0
    time.sleep(1.5)  # producer would finish instantly without the gate
    with open(marker) as f:
        produced = int(f.read())
    assert produced <= 1 + 64 + 4, produced  # cap + slack, far below 200

    # drain: everything arrives, in order
    rest = [ray.get(r, timeout=60) for r in it]
    assert rest == list(range(1, n))


def test_borrow_protocol_defers_free(ray_start_regular):
    """The owner defers freeing an object while a borrower (an actor
    holding the ref) is alive (reference: WaitForRefRemoved)."""
    import gc

    from ray_amd._core import runtime as rtmod

    @ray.remote
    class Holder:
        def hold(self, refs):
            self.ref = refs[0]  # nested -> stays a ref (borrowed)
            return True

        def read(self):
            return int(ray.get(self.ref)[123])

        def drop(self):
            del self.ref
            import gc as _gc

            _gc.collect()
            return True

    h = Holder.remote()
    big = ray.put(np.arange(300_000, dtype=np.int64))
    oid = big.id
    assert ray.get(h.hold.remote([big]), timeout=30)
    time.sleep(0.3)  # let the borrow_add notify land

    rt = rtmod.global_runtime()
    del big
    gc.collect()
    time.sleep(0.3)
    # owner count dropped, but the borrow defers the free
    assert oid in rt._pending_free or rt._borrows.get(oid, 0) > 0
    assert ray.get(h.read.remote(), timeout=30) == 123  # still readable

    assert ray.get(h.drop.remote(), timeout=30)
    deadline = time.time() + 10
    while time.time() < deadline and oid in rt.memory_store:
        time.sleep(0.1)
    assert oid not in rt.memory_store  # freed after the last release


def test_accelerator_type_and_timeline(ray_start_regular, tmp_path):
    from ray_amd import experimental as exp
    from ray_amd.util.accelerators import AMD_INSTINCT_MI355X

    # accelerator_type maps to the node's accelerator resource
    exp.set_resource(f"accelerator_type:{AMD_INSTINCT_MI355X}", 1)
    time.sleep(0.3)

    @ray.remote(accelerator_type=AMD_INSTINCT_MI355X)
    def on_mi355x():
        return "scheduled"

    assert ray.get(on_mi355x.remote(), timeout=30) == "scheduled"

    # ray.timeline: chrome-trace events incl. the task above (events
    # batch through raylet -> GCS; poll briefly)
    deadline = time.time() + 15
    evs = []
    while time.time() < deadline:
        evs = ray.timeline()
        if any(e["name"] == "on_mi355x" for e in evs):
            break
        time.sleep(0.3)
    assert any(e["name"] == "on_mi355x" for e in evs)
    out = tmp_path / "tl.json"
    ray.timeline(str(out))
    import json

    assert json.loads(out.read_text())


def test_pubsub_multi_subscriber_fanout(ray_start_regular):
    from ray_amd.util import pubsub

    with pubsub.Subscriber("fan") as a, pubsub.Subscriber("fan") as b:
        # one GCS connection per process: publish counts conns, every
        # local subscriber still receives
        assert pubsub.publish("fan", 7) >= 1
        assert a.poll(timeout=10) == 7
        assert b.poll(timeout=10) == 7


def test_pin_while_mapped_blocks_recycle(ray_start_regular):
    """Round-1 known limit, now closed: a consumer holding a ZERO-COPY
    view of a sealed segment pins it at the raylet; the owner's free
    must not recycle the segment into the hot pool (a later put of the
    same size class would overwrite the consumer's live view)."""

    @ray.remote
    class Viewer:
        def __init__(self):
            self.view = None

        def hold_view(self, ref_box):
            # get() maps the segment; keep the zero-copy numpy view
            self.view = ray.get(ref_box[0])
            return float(self.view[:100].sum())

        def re_read(self):
            return float(self.view[:100].sum())

    v = Viewer.remote()
    size = 512 * 1024  # one pool size-class exactly
    a = np.arange(size // 8, dtype=np.float64)
    ref = ray.put(a)
    want = float(a[:100].sum())
    assert ray.get(v.hold_view.remote([ref]), timeout=60) == want
    # owner drops its ref -> free path runs; the viewer actor dropped
    # its REF (ref_box was transient) but still holds the mapped VIEW
    del ref, a
    import gc

    gc.collect()
    time.sleep(1.0)
    # hammer same-size-class puts: with the race, one of these would
    # recycle the viewer's segment and overwrite its bytes
    spam = [ray.put(np.full(size // 8, 7.0)) for _ in range(8)]
    assert ray.get(v.re_read.remote(), timeout=60) == want
    del spam


def test_handler_event_loop_stats(ray_start_regular):
    """Per-handler event-loop stats (reference: event_stats.cc)."""
    @ray.remote
    def f():
        return 1

    ray.get([f.remote() for _ in range(5)])
    from ray_amd.util.state import node_debug_state

    st = node_debug_state()
    hs = st["raylet"]["handler_stats"]
    assert "request_lease" in hs and hs["request_lease"]["count"] >= 1
    assert hs["request_lease"]["mean_us"] > 0
    assert st["gcs"]["handler_stats"]["report_resources"]["count"] >= 1
    assert st["raylet"]["store"]["capacity"] > 0


def test_dashboard_routes(ray_start_regular):
    """Dashboard REST surface (reference: dashboard/modules/*):
    nodes/actors/objects/debug_state/logs all serve JSON."""
    import asyncio
    import json as _json

    from ray_amd.dashboard import build_asgi_app

    @ray.remote
    def f():
        return 1

    ray.get(f.remote())
    app = build_asgi_app()

    async def hit(path):
        out = {}

        async def send(msg):
            if msg["type"] == "http.response.start":
                out["status"] = msg["status"]
            else:
                out.setdefault("body", b"")
                out["body"] += msg.get("body", b"")

        async def receive():
            return {"type": "http.request"}

        await app({"type": "http", "path": path, "headers": []},
                  receive, send)
        return out

    loop = asyncio.new_event_loop()
    try:
        for p in ("/api/nodes", "/api/actors", "/api/objects",
                  "/api/cluster_status", "/api/debug_state", "/api/logs",
                  "/api/serve", "/api/placement_groups"):
            r = loop.run_until_complete(hit(p))
            assert r["status"] == 200, (p, r)
            _json.loads(r["body"])
        r = loop.run_until_complete(hit("/metrics"))
        assert r["status"] == 200
        r = loop.run_until_complete(hit("/"))
        assert r["status"] == 200 and b"ray_amd" in r["body"]
    finally:
        loop.close()


def test_usage_stats_local_report():
    """Usage stats (reference: _private/usage): opt-out flag, library
    tagging, and a LOCAL-ONLY report written at shutdown (no egress in
    this build)."""
    import json as _json
    import os as _os

    from ray_amd.util import usage_stats as us

    _os.environ["RAY_AMD_USAGE_STATS_ENABLED"] = "1"
    try:
        assert us.usage_stats_enabled()
        us.record_library_usage("data")
        us.record_extra_usage_tag("test", "1")
        ctx = ray.init(num_cpus=2, ignore_reinit_error=True)
        session_dir = ctx.session_dir
        ray.shutdown()
        p = _os.path.join(session_dir, "usage_stats.json")
        # session dir is cleaned at shutdown by the owning driver, so
        # validate the generator directly instead when it's gone
        rep = us.generate_report()
        assert "data" in rep["library_usages"]
        assert rep["extra_usage_tags"]["test"] == "1"
        _os.environ["RAY_AMD_USAGE_STATS_ENABLED"] = "0"
        assert not us.usage_stats_enabled()
    finally:
        _os.environ.pop("RAY_AMD_USAGE_STATS_ENABLED", None)


def test_tracing_spans_cross_process():
    """Spans cross process boundaries (reference:
    util/tracing/tracing_helper.py:183-193): a driver span's trace_id
    is inherited by the remote task's execution span, and a span
    opened INSIDE the task nests under it."""
    import os as _os

    _os.environ["RAY_AMD_TRACING"] = "1"
    try:
        ray.init(num_cpus=2, ignore_reinit_error=True)
        from ray_amd.util.tracing import get_trace_events, span

        @ray.remote
        def traced_task():
            from ray_amd.util.tracing import span as span2

            with span2("inner-work"):
                return 42

        with span("driver-root") as root:
            assert ray.get(traced_task.remote()) == 42
        import time as _t

        deadline = _t.time() + 10
        spans = []
        while _t.time() < deadline:
            spans = get_trace_events()
            names = {s["name"] for s in spans
                     if s["trace_id"] == root.trace_id}
            if {"driver-root", "task:traced_task", "inner-work"} <= names:
                break
            _t.sleep(0.2)
        by_name = {s["name"]: s for s in spans
                   if s["trace_id"] == root.trace_id}
        assert "task:traced_task" in by_name, by_name.keys()
        assert by_name["task:traced_task"]["parent_id"] == root.span_id
        assert "inner-work" in by_name
        assert (by_name["inner-work"]["parent_id"]
                == by_name["task:traced_task"]["span_id"])
    finally:
        _os.environ.pop("RAY_AMD_TRACING", None)
        ray.shutdown()


def test_experimental_shuffle(ray_start_regular):
    """Push-based shuffle prototype (reference:
    experimental/shuffle.py): mappers push partitions to reducer
    actors; outputs partition by key."""
    from ray_amd.experimental.shuffle import shuffle

    blocks = [list(range(i * 10, (i + 1) * 10)) for i in range(4)]

    def partition(block, n):
        outs = [[] for _ in range(n)]
        for x in block:
            outs[x % n].append(x)
        return outs

    out = shuffle(blocks, 3, partition, lambda ps: sorted(sum(ps, [])))
    assert sorted(sum(out, [])) == list(range(40))
    for i, part in enumerate(out):
        assert all(x % 3 == i for x in part)
