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


def test_tensor_transport_option_plumbs(ray_start_regular):
    """tensor_transport actors (reference: actor.py:621): CPU tensors
    pass through unchanged (offload only triggers for CUDA tensors);
    the option survives handle pickling."""

    @ray.remote(tensor_transport="hipipc")
    class T:
        def make(self, n):
            return torch.arange(n, dtype=torch.float32)

        def double(self, t):
            return t * 2

    a = T.remote()
    t = ray.get(a.make.remote(8))
    assert torch.equal(t, torch.arange(8, dtype=torch.float32))
    t2 = ray.get(a.double.remote(t))
    assert float(t2.sum()) == 2 * float(t.sum())
    import cloudpickle

    a2 = cloudpickle.loads(cloudpickle.dumps(a))
    assert a2._tensor_transport == "hipipc"


@pytest.mark.gpu
def test_tensor_transport_zero_copy_gpu():
    """GPU tensors returned by a tensor_transport actor reach the
    consumer as hipIpc views: mutating the producer's tensor AFTER the
    consumer fetched it is visible to the consumer (proof the storage
    is shared, not copied)."""
    ray.init(num_cpus=4, num_gpus=1, ignore_reinit_error=True)
    try:
        @ray.remote(num_gpus=0.3, tensor_transport="hipipc")
        class Prod:
            def __init__(self):
                self.t = None

            def make(self, n):
                self.t = torch.ones(n, device="cuda")
                return self.t

            def mutate(self):
                self.t += 41  # in-place: consumers sharing storage see it
                torch.cuda.synchronize()
                return True

        @ray.remote(num_gpus=0.3)
        class Cons:
            def __init__(self):
                self.got = None

            def recv(self, t):
                self.got = t  # auto-fetched hipIpc view
                return float(t.sum())

            def used_fallback(self):
                from ray_amd.experimental.rdt import get_gpu_object_store

                return get_gpu_object_store().last_fetch_fallback

            def re_read(self):
                torch.cuda.synchronize()
                return float(self.got.sum())

        p = Prod.remote()
        c = Cons.remote()
        n = 1024
        t_ref = p.make.remote(n)
        # data correctness holds ALWAYS (checksum-verified mapping
        # with staged fallback)
        assert ray.get(c.recv.remote(t_ref), timeout=120) == float(n)
        ray.get(p.mutate.remote(), timeout=60)
        if ray.get(c.used_fallback.remote(), timeout=60):
            # platform dmabuf mapping faulted this run (rare; see
            # rdt.py checksum note) — the consumer holds a verified
            # COPY, so shared-storage semantics don't apply
            assert ray.get(c.re_read.remote(), timeout=60) == float(n)
        else:
            # shared storage: consumer's view reflects the mutation
            assert ray.get(c.re_read.remote(), timeout=60) == float(n * 42)
    finally:
        ray.shutdown()


@pytest.mark.gpu
def test_dag_gpu_channel_hipipc():
    """Compiled-DAG edges carrying GPU tensors ship hipIpc refs (no
    host staging): a two-actor GPU pipeline through shm channels."""
    ray.init(num_cpus=4, num_gpus=1, ignore_reinit_error=True)
    try:
        from ray_amd.dag import InputNode

        @ray.remote(num_gpus=0.3)
        class Stage1:
            def f(self, x):
                return torch.full((256,), float(x), device="cuda")

        @ray.remote(num_gpus=0.3)
        class Stage2:
            def g(self, t):
                return float((t * 2).sum().item())

        a = Stage1.bind()
        b = Stage2.bind()
        with InputNode() as inp:
            mid = a.f.bind(inp)
            out = b.g.bind(mid)
        compiled = out.experimental_compile()
        try:
            for i in range(5):
                fut = compiled.execute(i)
                assert fut.get(timeout=120) == 256.0 * i * 2
        finally:
            compiled.teardown()
    finally:
        ray.shutdown()
