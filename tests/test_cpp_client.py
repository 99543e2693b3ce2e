"""C++ client API (reference: cpp/ C++ worker API) — compiles the
demo binary against csrc/cpp_client/ray_client.hpp and drives a live
cluster through it: KV, node table, registered tasks, named-actor
calls, pub/sub publish, error surfacing."""
import os
import subprocess
import sys

import pytest

import ray_amd as ray

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SRC = os.path.join(ROOT, "ray_amd", "csrc", "cpp_client")


@pytest.fixture(scope="module")
def demo_bin(tmp_path_factory):
    out = str(tmp_path_factory.mktemp("cpp") / "demo")
    r = subprocess.run(
        ["g++", "-std=c++17", "-O2", "-o", out,
         os.path.join(SRC, "demo.cc")],
        capture_output=True, text=True,
    )
    assert r.returncode == 0, r.stderr
    return out


def test_cpp_client_end_to_end(demo_bin):
    from ray_amd.client.server import ClientServer
    from ray_amd.util import pubsub

    ray.init(num_cpus=2, ignore_reinit_error=True)
    try:
        @ray.remote
        def add(a, b):
            return a + b

        @ray.remote
        class Counter:
            def __init__(self):
                self.v = 0

            def incr(self, n):
                self.v += n
                return self.v

        Counter.options(name="counter").remote()
        srv = ClientServer(port=0)
        srv.register_task("add", add)
        port = srv.start()

        with pubsub.Subscriber("cpp_events") as sub:
            r = subprocess.run([demo_bin, str(port)], capture_output=True,
                               text=True, timeout=120)
            assert r.returncode == 0, r.stdout + r.stderr
            lines = dict(
                l.split(": ", 1) for l in r.stdout.strip().splitlines()
            )
            assert lines["kv"] == "cpp_value"
            assert int(lines["nodes"]) == 1
            assert lines["task add"] == "42"
            assert lines["actor"] == "5 12"
            assert int(lines["published"]) >= 1
            assert lines["error"] == "caught"
            assert sub.poll(timeout=10) == "hello-from-cpp"

        # python sees the C++ KV write
        from ray_amd.experimental import internal_kv

        assert internal_kv._internal_kv_get(b"cpp_key") == b"cpp_value"
    finally:
        ray.shutdown()
