"""Placement group + multi-node scheduling tests (reference model:
python/ray/tests/test_placement_group*.py)."""
import pytest

import ray_amd as ray
from ray_amd.util import (
    PlacementGroupSchedulingStrategy,
    placement_group,
    placement_group_table,
    remove_placement_group,
)


def test_pg_pack_create_and_use(ray_start_regular):
    pg = placement_group([{"CPU": 1}, {"CPU": 1}], strategy="PACK")
    assert pg.wait(30)

    @ray.remote
    def where():
        import os

        return os.getpid()

    r = ray.get(
        where.options(
            scheduling_strategy=PlacementGroupSchedulingStrategy(
                placement_group=pg, placement_group_bundle_index=0
            ),
            num_cpus=1,
        ).remote()
    )
    assert isinstance(r, int)
    remove_placement_group(pg)


def test_pg_table(ray_start_regular):
    pg = placement_group([{"CPU": 1}], strategy="PACK", name="mypg")
    assert pg.wait(30)
    table = placement_group_table()
    ent = table[pg.id.hex()]
    assert ent["state"] == "CREATED"
    assert ent["name"] == "mypg"
    remove_placement_group(pg)


def test_pg_infeasible_strict_spread():
    import ray_amd as ray

    ray.init(num_cpus=2, ignore_reinit_error=True)
    try:
        # single node: STRICT_SPREAD of 2 bundles cannot be placed
        pg = placement_group([{"CPU": 1}, {"CPU": 1}], strategy="STRICT_SPREAD")
        assert not pg.wait(2)
    finally:
        ray.shutdown()


def test_pg_actor_placement(ray_start_regular):
    pg = placement_group([{"CPU": 2}], strategy="PACK")
    assert pg.wait(30)

    @ray.remote(num_cpus=1)
    class A:
        def ping(self):
            return "pong"

    a = A.options(
        scheduling_strategy=PlacementGroupSchedulingStrategy(
            placement_group=pg, placement_group_bundle_index=0
        )
    ).remote()
    assert ray.get(a.ping.remote()) == "pong"
    remove_placement_group(pg)


def test_multi_node_cluster(ray_start_cluster):
    cluster = ray_start_cluster
    cluster.add_node(num_cpus=2, resources={"special": 2})
    cluster.connect()
    cluster.wait_for_nodes()
    assert len([n for n in ray.nodes() if n["Alive"]]) == 2
    total = ray.cluster_resources()
    assert total["CPU"] == 6
    assert total.get("special") == 2

    # task requiring the remote node's custom resource spills over
    @ray.remote(resources={"special": 1}, num_cpus=1)
    def on_special():
        return "ran"

    assert ray.get(on_special.remote(), timeout=60) == "ran"


def test_multi_node_object_transfer(ray_start_cluster):
    import numpy as np

    cluster = ray_start_cluster
    cluster.add_node(num_cpus=2, resources={"remote_node": 2})
    cluster.connect()
    cluster.wait_for_nodes()

    @ray.remote(resources={"remote_node": 1}, num_cpus=1)
    def produce():
        return np.ones((2000, 2000))  # 32 MB -> shm store on remote node

    @ray.remote(num_cpus=1)
    def consume(a):
        return float(a.sum())

    ref = produce.remote()
    # driver pulls the object across "nodes"
    a = ray.get(ref, timeout=120)
    assert a.shape == (2000, 2000)
    # and a task on the head node can consume it as an arg
    assert ray.get(consume.remote(ref), timeout=120) == 4000000.0


def test_strict_spread_on_two_nodes(ray_start_cluster):
    cluster = ray_start_cluster
    cluster.add_node(num_cpus=4)
    cluster.connect()
    cluster.wait_for_nodes()
    pg = placement_group([{"CPU": 1}, {"CPU": 1}], strategy="STRICT_SPREAD")
    assert pg.wait(30)
    table = placement_group_table()
    nodes = table[pg.id.hex()]["bundle_nodes"]
    assert nodes[0] != nodes[1]
    remove_placement_group(pg)
