"""Ray-Client mode: thin client driving a remote cluster over TCP
(reference: python/ray/util/client/)."""
import os
import subprocess
import sys
import time

import numpy as np
import pytest

import ray_amd as ray

SERVER_CODE = """
import sys
import time

import ray_amd as ray
from ray_amd.client.server import ClientServer

ray.init(num_cpus=4)
port = ClientServer(port=0).start()
with open(sys.argv[1] + ".tmp", "w") as f:
    f.write(str(port))
import os
os.replace(sys.argv[1] + ".tmp", sys.argv[1])
time.sleep(600)
"""


@pytest.fixture
def client_server(tmp_path):
    port_file = str(tmp_path / "port")
    proc = subprocess.Popen(
        [sys.executable, "-c", SERVER_CODE, port_file],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    )
    try:
        deadline = time.time() + 60
        while not os.path.exists(port_file):
            assert proc.poll() is None, "client server died"
            assert time.time() < deadline, "client server startup timeout"
            time.sleep(0.1)
        with open(port_file) as f:
            port = int(f.read())
        yield port
    finally:
        proc.kill()
        proc.wait(10)


def test_client_tasks_objects_actors(client_server):
    ctx = ray.init(f"ray_amd://127.0.0.1:{client_server}")
    try:
        assert ray.is_initialized()

        # tasks + nested refs in args
        @ray.remote
        def double(x):
            return x * 2

        assert ray.get(double.remote(21), timeout=60) == 42
        inner = double.remote(10)
        assert ray.get(double.remote(inner), timeout=60) == 40

        # put / get / wait with numpy payloads
        ref = ray.put(np.arange(1000))
        out = ray.get(ref, timeout=60)
        assert out[999] == 999
        ready, rest = ray.wait([ref], timeout=30)
        assert ready and not rest

        # error surfacing
        @ray.remote
        def boom():
            raise ValueError("kapow")

        with pytest.raises(ray.exceptions.RayTaskError):
            ray.get(boom.remote(), timeout=60)

        # actors incl. named lookup and kill
        @ray.remote
        class Counter:
            def __init__(self, start):
                self.v = start

            def incr(self, n=1):
                self.v += n
                return self.v

        c = Counter.options(name="client_counter").remote(100)
        assert ray.get(c.incr.remote(), timeout=60) == 101
        c2 = ray.get_actor("client_counter")
        assert ray.get(c2.incr.remote(5), timeout=60) == 106
        ray.kill(c)

        # cluster info passthrough
        assert ray.cluster_resources().get("CPU", 0) >= 4
        assert len(ray.nodes()) == 1
    finally:
        ray.shutdown()


def test_client_mode_contracts(client_server):
    """Client-mode edge contracts: streaming is explicitly unsupported,
    runtime_env env_vars pass through, working_dir staging errors
    clearly."""
    ctx = ray.init(f"ray_amd://127.0.0.1:{client_server}")
    try:
        @ray.remote(num_returns="streaming")
        def gen():
            yield 1

        with pytest.raises(NotImplementedError):
            gen.remote()

        @ray.remote
        def read_env():
            import os

            return os.environ.get("CLIENT_FLAG")

        # env_vars runtime_env works over the client
        r = read_env.options(
            runtime_env={"env_vars": {"CLIENT_FLAG": "on"}}
        ).remote()
        assert ray.get(r, timeout=60) == "on"

        with pytest.raises(NotImplementedError):
            read_env.options(
                runtime_env={"working_dir": "."}
            ).remote()
    finally:
        ray.shutdown()
