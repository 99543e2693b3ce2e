"""Two-"machine" TCP harness: every daemon on TCP (127.0.0.1), each
node with its own shm dir — proves no hidden same-filesystem
assumptions (object pulls must move bytes through the chunked raylet
transfer). Reference: network-addressed startup, _private/node.py:1422.
"""
import os
import time

import numpy as np
import pytest

import ray_amd as ray


@pytest.fixture
def tcp_cluster():
    from ray_amd.cluster_utils import Cluster

    prev_ip = os.environ.get("RAY_AMD_NODE_IP")
    cluster = Cluster(tcp=True)
    yield cluster
    cluster.shutdown()
    if prev_ip is None:
        os.environ.pop("RAY_AMD_NODE_IP", None)
    else:
        os.environ["RAY_AMD_NODE_IP"] = prev_ip


def test_tcp_cluster_tasks_and_objects(tcp_cluster):
    cluster = tcp_cluster
    assert cluster.gcs_addr.startswith("tcp:")
    n2 = cluster.add_node(num_cpus=2, resources={"n2": 1})
    assert n2.addr.startswith("tcp:")
    cluster.connect()
    cluster.wait_for_nodes()

    @ray.remote(resources={"n2": 0.5})
    def on_n2():
        # a big payload created on node 2's OWN shm dir
        return np.arange(600_000, dtype=np.float64)

    ref = on_n2.remote()
    val = ray.get(ref, timeout=90)
    assert val[123456] == 123456  # bytes crossed node shm dirs

    # and the reverse direction: head-created object consumed on n2
    big = ray.put(np.full(400_000, 7, dtype=np.int64))

    @ray.remote(resources={"n2": 0.5})
    def consume(x):
        return int(x.sum())

    assert ray.get(consume.remote(big), timeout=90) == 7 * 400_000


def test_tcp_cluster_actors_cross_node(tcp_cluster):
    cluster = tcp_cluster
    cluster.add_node(num_cpus=2, resources={"n2": 1})
    cluster.connect()
    cluster.wait_for_nodes()

    @ray.remote(resources={"n2": 0.5})
    class A:
        def where(self):
            return os.environ.get("RAY_AMD_SHM_DIR", "")

        def echo(self, x):
            return x

    a = A.remote()
    shm = ray.get(a.where.remote(), timeout=60)
    assert shm.endswith("_node1")  # actor really on the other "machine"
    assert ray.get(a.echo.remote({"k": [1, 2, 3]}), timeout=60) == {"k": [1, 2, 3]}


def test_tcp_node_death(tcp_cluster):
    cluster = tcp_cluster
    n2 = cluster.add_node(num_cpus=2, resources={"n2": 1})
    cluster.connect()
    cluster.wait_for_nodes()

    @ray.remote(resources={"n2": 0.5})
    class OnN2:
        def ping(self):
            return "pong"

    a = OnN2.remote()
    assert ray.get(a.ping.remote(), timeout=60) == "pong"
    cluster.remove_node(n2)
    time.sleep(1.0)
    with pytest.raises(ray.exceptions.RayError):
        ray.get(a.ping.remote(), timeout=30)
    alive = [n for n in ray.nodes() if n["Alive"]]
    assert len(alive) == 1


def test_pull_dedup_and_priority(tcp_cluster):
    """Pull manager: concurrent pulls of one object share a single
    transfer (dedup, reference pull_manager.h), and the push-streamed
    path moves large objects across separate shm dirs correctly."""
    cluster = tcp_cluster
    cluster.add_node(num_cpus=4, resources={"n2": 1})
    cluster.connect()
    cluster.wait_for_nodes()

    @ray.remote(resources={"n2": 0.1})
    def make_big():
        return np.arange(2_000_000, dtype=np.float64)  # 16 MB, 4 chunks+

    ref = make_big.remote()
    ray.wait([ref], timeout=60)

    # many concurrent consumers on the head node pulling the SAME oid
    @ray.remote
    def consume(x, i):
        return float(x[i])

    outs = ray.get([consume.remote(ref, i) for i in range(8)], timeout=120)
    assert outs == [float(i) for i in range(8)]


def test_wait_fetch_local_pulls_payload(tcp_cluster):
    """ray.wait(fetch_local=True) starts the cross-node transfer at
    WAIT priority: after wait readies the ref, the payload is already
    in THIS node's shm dir (reference: ray.wait fetch_local
    semantics)."""
    cluster = tcp_cluster
    cluster.add_node(num_cpus=2, resources={"n2": 1})
    cluster.connect()
    cluster.wait_for_nodes()

    @ray.remote(resources={"n2": 0.5})
    def big_on_n2():
        return np.arange(700_000, dtype=np.float64)

    ref = big_on_n2.remote()
    ready, _ = ray.wait([ref], timeout=60, fetch_local=True)
    assert ready == [ref]
    rt = ray.api._rt.global_runtime()
    path = os.path.join(rt.shm_dir, ref.id.hex())
    deadline = time.time() + 10
    while not os.path.exists(path) and time.time() < deadline:
        time.sleep(0.05)
    assert os.path.exists(path), "payload not pulled to the local node"
    got = ray.get(ref, timeout=30)
    assert got.shape == (700_000,)
