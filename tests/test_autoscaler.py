"""LocalAutoscaler tests (reference model: autoscaler tests with the
fake multi-node provider)."""
import time

import ray_amd as ray


def test_autoscaler_scales_up_on_demand(ray_start_cluster):
    cluster = ray_start_cluster  # head: 4 CPUs
    cluster.connect()
    from ray_amd.autoscaler import LocalAutoscaler

    asc = LocalAutoscaler(
        cluster, worker_resources={"CPU": 4}, max_workers=2,
        upscale_after_s=0.3,
    ).start()
    try:
        @ray.remote(num_cpus=1)
        def slow():
            time.sleep(2)
            return 1

        # 12 one-cpu tasks on a 4-cpu head: queue builds -> scale up
        refs = [slow.remote() for _ in range(12)]
        deadline = time.time() + 30
        while time.time() < deadline and asc.num_workers == 0:
            time.sleep(0.2)
        assert asc.num_workers >= 1, "autoscaler did not add a node"
        assert ray.get(refs, timeout=120) == [1] * 12
        assert len([n for n in ray.nodes() if n["Alive"]]) >= 2
    finally:
        asc.stop()


def test_request_resources_triggers_scale(ray_start_cluster):
    cluster = ray_start_cluster
    cluster.connect()
    from ray_amd.autoscaler import LocalAutoscaler, sdk

    asc = LocalAutoscaler(
        cluster, worker_resources={"CPU": 8}, max_workers=1,
        upscale_after_s=0.2,
    ).start()
    try:
        sdk.request_resources(num_cpus=10)  # head has only 4
        deadline = time.time() + 20
        while time.time() < deadline and asc.num_workers == 0:
            time.sleep(0.2)
        assert asc.num_workers == 1
        assert ray.cluster_resources()["CPU"] >= 10
    finally:
        asc.stop()
