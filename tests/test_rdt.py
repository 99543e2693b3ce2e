"""RDT GPU object store tests (reference model: python/ray/tests/rdt/).

CPU tests exercise the staged path; the gpu-marked test exercises the
hipIpc zero-copy path with two actors sharing one MI355X.
"""
import numpy as np
import pytest
import torch

import ray_amd as ray


@ray.remote
class Producer:
    def __init__(self, device="cpu"):
        self.device = device
        from ray_amd.experimental import get_gpu_object_store

        self.store = get_gpu_object_store()

    def make(self, n):
        t = torch.arange(n, dtype=torch.float32, device=self.device) * 2.0
        return self.store.put(t)

    def free(self, ref):
        self.store.free(ref)

    def num_objects(self):
        return self.store.num_objects()


@ray.remote
class Consumer:
    def __init__(self, device="cpu"):
        self.device = device
        from ray_amd.experimental import get_gpu_object_store

        self.store = get_gpu_object_store()

    def total(self, ref):
        t = self.store.get(ref, device=self.device)
        return float(t.sum())

    def ptr_and_total(self, ref):
        t = self.store.get(ref)
        return t.data_ptr(), float(t.sum())


def test_rdt_staged_cpu(ray_start_regular):
    p = Producer.remote()
    c = Consumer.remote()
    ref = ray.get(p.make.remote(100))
    assert ray.get(c.total.remote(ref)) == float(np.arange(100).sum() * 2)
    assert ray.get(p.num_objects.remote()) == 1
    ray.get(p.free.remote(ref))
    assert ray.get(p.num_objects.remote()) == 0


def test_rdt_ref_is_small(ray_start_regular):
    import cloudpickle

    p = Producer.remote()
    ref = ray.get(p.make.remote(1000000))
    assert len(cloudpickle.dumps(ref)) < 500  # handle, not data


@pytest.mark.gpu
def test_rdt_hipipc_zero_copy():
    """Two actors pinned to the same MI355X exchange a tensor by
    hipIpcMemHandle (no host copy)."""
    assert torch.cuda.is_available()
    ray.init(num_gpus=1, num_cpus=4, ignore_reinit_error=True)
    try:
        p = Producer.options(num_gpus=0.5).remote("cuda")
        c = Consumer.options(num_gpus=0.5).remote("cuda")
        ref = ray.get(p.make.remote(1 << 20))
        expected = float(torch.arange(1 << 20, dtype=torch.float32).sum() * 2)
        got = ray.get(c.total.remote(ref), timeout=120)
        assert got == expected
        # repeated get hits consumer cache and stays consistent
        got2 = ray.get(c.total.remote(ref), timeout=120)
        assert got2 == expected
    finally:
        ray.shutdown()
