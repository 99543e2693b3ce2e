"""ray.util.collective tests over actors, gloo backend (CPU).

Pattern from reference: comm-layer logic testable without GPUs
(fake/CPU communicator). The same code path with
backend="rccl" runs over RCCL/xGMI on the GPU box (test_collective_gpu)."""
import pytest
import torch

import ray_amd as ray


@ray.remote
class Worker:
    def __init__(self, rank, world_size, backend="torch_gloo", group="g1"):
        from ray_amd.util import collective as col

        self.col = col
        self.rank = rank
        self.group = group
        col.init_collective_group(world_size, rank, backend=backend,
                                  group_name=group)

    def do_allreduce(self):
        t = torch.ones(4) * (self.rank + 1)
        self.col.allreduce(t, self.group)
        return t.tolist()

    def do_broadcast(self):
        t = torch.full((3,), float(self.rank))
        self.col.broadcast(t, src_rank=1, group_name=self.group)
        return t.tolist()

    def do_allgather(self):
        out = [torch.zeros(2) for _ in range(2)]
        t = torch.full((2,), float(self.rank))
        self.col.allgather(out, t, self.group)
        return [o.tolist() for o in out]

    def do_reducescatter(self):
        parts = [torch.full((2,), float(self.rank + 1)) for _ in range(2)]
        out = torch.zeros(2)
        self.col.reducescatter(out, parts, self.group)
        return out.tolist()

    def do_sendrecv(self):
        if self.rank == 0:
            t = torch.tensor([41.0])
            self.col.send(t, 1, self.group)
            return t.tolist()
        t = torch.zeros(1)
        self.col.recv(t, 0, self.group)
        return t.tolist()

    def do_barrier(self):
        self.col.barrier(self.group)
        return self.col.get_rank(self.group)


@pytest.fixture
def two_workers(ray_start_regular):
    w0 = Worker.remote(0, 2)
    w1 = Worker.remote(1, 2)
    # wait for both inits (rendezvous)
    yield w0, w1


def test_allreduce(two_workers):
    w0, w1 = two_workers
    r0, r1 = ray.get([w0.do_allreduce.remote(), w1.do_allreduce.remote()])
    assert r0 == [3.0] * 4 and r1 == [3.0] * 4


def test_broadcast(two_workers):
    w0, w1 = two_workers
    r0, r1 = ray.get([w0.do_broadcast.remote(), w1.do_broadcast.remote()])
    assert r0 == [1.0, 1.0, 1.0] and r1 == [1.0, 1.0, 1.0]


def test_allgather(two_workers):
    w0, w1 = two_workers
    r0, r1 = ray.get([w0.do_allgather.remote(), w1.do_allgather.remote()])
    assert r0 == [[0.0, 0.0], [1.0, 1.0]]
    assert r1 == [[0.0, 0.0], [1.0, 1.0]]


def test_reducescatter(two_workers):
    w0, w1 = two_workers
    r0, r1 = ray.get(
        [w0.do_reducescatter.remote(), w1.do_reducescatter.remote()]
    )
    assert r0 == [3.0, 3.0] and r1 == [3.0, 3.0]


def test_send_recv(two_workers):
    w0, w1 = two_workers
    r0, r1 = ray.get([w0.do_sendrecv.remote(), w1.do_sendrecv.remote()])
    assert r1 == [41.0]


def test_barrier_and_rank(two_workers):
    w0, w1 = two_workers
    r = ray.get([w0.do_barrier.remote(), w1.do_barrier.remote()])
    assert sorted(r) == [0, 1]
