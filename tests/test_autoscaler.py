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


def test_autoscaler_v2_scales_for_demand_shapes():
    """v2: pending lease SHAPES drive bin-packed launches through the
    instance-manager FSM; idle nodes terminate after the timeout
    (reference: autoscaler/v2 scheduler + instance_manager)."""
    import ray_amd as ray
    from ray_amd.autoscaler import AutoscalerV2, LocalNodeProvider, NodeType
    from ray_amd.cluster_utils import Cluster

    cluster = Cluster(head_node_args={"num_cpus": 1})
    try:
        cluster.connect()
        asc = AutoscalerV2(
            LocalNodeProvider(cluster),
            [NodeType("worker_2cpu", {"CPU": 2}, min_workers=0,
                      max_workers=3)],
            idle_timeout_s=2.0, poll_s=0.2, upscale_after_s=0.2,
        ).start()
        try:
            @ray.remote(num_cpus=2)
            def heavy(i):
                time.sleep(1.0)
                return i

            # head has 1 CPU: every task needs a new 2-CPU worker shape
            refs = [heavy.remote(i) for i in range(4)]
            assert sorted(ray.get(refs, timeout=120)) == [0, 1, 2, 3]
            summ = asc.summary()
            assert summ.get("worker_2cpu", {}).get("RUNNING", 0) >= 1
            # idle scale-down
            deadline = time.time() + 40
            while time.time() < deadline:
                s = asc.summary().get("worker_2cpu", {})
                if s.get("RUNNING", 0) == 0 and s.get("TERMINATED", 0) >= 1:
                    break
                time.sleep(0.3)
            s = asc.summary().get("worker_2cpu", {})
            assert s.get("TERMINATED", 0) >= 1, s
        finally:
            asc.stop()
    finally:
        cluster.shutdown()
