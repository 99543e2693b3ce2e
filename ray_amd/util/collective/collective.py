"""Declarative collective groups over actors (reference:
python/ray/util/collective/collective.py:149-624).

MI355X-native backend: torch.distributed process-group objects built
directly on a TCPStore — backend "rccl" is torch's ProcessGroupNCCL,
which IS RCCL on ROCm (collectives run over xGMI in-node); "torch_gloo"
is the CPU path. Multiple named groups can coexist per process (the
reference's cupy-NCCL group cache, nccl_collective_group.py:126), and
bf16 IS supported (the reference notes cupy couldn't, nccl_util.py:693).

Rendezvous runs through the GCS KV: rank 0 publishes host:port under
"collective:<group>"; everyone else polls. Outside a ray_amd session
(e.g. plain torchrun) MASTER_ADDR/MASTER_PORT env are used.
"""
from __future__ import annotations

import datetime
import os
import socket
import time
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from .types import Backend, ReduceOp, torch_reduce_op

_groups: Dict[str, "Group"] = {}


class Group:
    def __init__(self, pg, rank: int, world_size: int, backend: str, store=None):
        self.pg = pg
        self.rank = rank
        self.world_size = world_size
        self.backend = backend
        self._store = store  # keep TCPStore alive


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _kv_rendezvous(group_name: str, rank: int, timeout: float = 120.0) -> str:
    from ..._core import runtime as _rt

    rt = _rt.global_runtime()
    key = f"collective:{group_name}".encode()
    if rank == 0:
        addr = f"127.0.0.1:{_free_port()}"
        rt.gcs_call("kv_put", {"ns": "collective", "key": key,
                               "value": addr.encode()})
        return addr
    deadline = time.time() + timeout
    while time.time() < deadline:
        v = rt.gcs_call("kv_get", {"ns": "collective", "key": key})
        if v:
            return bytes(v).decode()
        time.sleep(0.02)
    raise TimeoutError(f"rendezvous for group {group_name} timed out")


def init_collective_group(
    world_size: int,
    rank: int,
    backend: str = Backend.RCCL,
    group_name: str = "default",
) -> None:
    """Join a named collective group from within an actor/task."""
    if group_name in _groups:
        raise RuntimeError(f"group {group_name} already initialized here")
    from ..._core import runtime as _rt

    if _rt.is_initialized():
        addr = _kv_rendezvous(group_name, rank)
        host, port = addr.split(":")
    else:
        host = os.environ.get("MASTER_ADDR", "127.0.0.1")
        port = os.environ.get("MASTER_PORT", "29500")
    store = dist.TCPStore(
        host, int(port), world_size, is_master=(rank == 0),
        timeout=datetime.timedelta(seconds=120),
    )
    if backend in ("rccl", "nccl"):
        opts = dist.ProcessGroupNCCL.Options()
        pg = dist.ProcessGroupNCCL(store, rank, world_size, opts)
        backend = "rccl"
    else:
        pg = dist.ProcessGroupGloo(store, rank, world_size)
        backend = "torch_gloo"
    _groups[group_name] = Group(pg, rank, world_size, backend, store)


def create_collective_group(
    actors: List,
    world_size: int,
    ranks: List[int],
    backend: str = Backend.RCCL,
    group_name: str = "default",
):
    """Declare a group for a set of actors (reference :186). The actual
    comm is initialized inside each actor via init_collective_group."""
    from ..._core import runtime as _rt

    rt = _rt.global_runtime()
    key = f"collective_decl:{group_name}".encode()
    import msgpack

    rt.gcs_call(
        "kv_put",
        {
            "ns": "collective",
            "key": key,
            "value": msgpack.packb(
                {"world_size": world_size, "ranks": ranks, "backend": backend}
            ),
        },
    )


def destroy_collective_group(group_name: str = "default"):
    g = _groups.pop(group_name, None)
    del g


def is_group_initialized(group_name: str = "default") -> bool:
    return group_name in _groups


def _get(group_name: str) -> Group:
    g = _groups.get(group_name)
    if g is None:
        raise RuntimeError(
            f"collective group {group_name!r} is not initialized; call "
            "init_collective_group() first"
        )
    return g


def get_rank(group_name: str = "default") -> int:
    return _get(group_name).rank


def get_collective_group_size(group_name: str = "default") -> int:
    return _get(group_name).world_size


def allreduce(tensor, group_name: str = "default", op: ReduceOp = ReduceOp.SUM):
    g = _get(group_name)
    opts = dist.AllreduceOptions()
    opts.reduceOp = torch_reduce_op(op)
    g.pg.allreduce([tensor], opts).wait()


def allreduce_multigpu(tensor_list, group_name="default", op=ReduceOp.SUM):
    for t in tensor_list:
        allreduce(t, group_name, op)


def reduce(tensor, dst_rank: int = 0, group_name: str = "default",
           op: ReduceOp = ReduceOp.SUM):
    g = _get(group_name)
    opts = dist.ReduceOptions()
    opts.rootRank = dst_rank
    opts.reduceOp = torch_reduce_op(op)
    g.pg.reduce([tensor], opts).wait()


def broadcast(tensor, src_rank: int = 0, group_name: str = "default"):
    g = _get(group_name)
    opts = dist.BroadcastOptions()
    opts.rootRank = src_rank
    g.pg.broadcast([tensor], opts).wait()


def allgather(tensor_list: List, tensor, group_name: str = "default"):
    g = _get(group_name)
    g.pg.allgather([tensor_list], [tensor]).wait()


def reducescatter(tensor, tensor_list: List, group_name: str = "default",
                  op: ReduceOp = ReduceOp.SUM):
    g = _get(group_name)
    if g.backend == "rccl":
        opts = dist.ReduceScatterOptions()
        opts.reduceOp = torch_reduce_op(op)
        g.pg.reduce_scatter([tensor], [tensor_list], opts).wait()
    else:
        # gloo lacks reduce_scatter: allreduce each shard then slice
        full = torch.cat([t.flatten() for t in tensor_list])
        g.pg.allreduce([full]).wait()
        n = tensor.numel()
        tensor.copy_(full[g.rank * n : (g.rank + 1) * n].view_as(tensor))


def barrier(group_name: str = "default"):
    g = _get(group_name)
    if hasattr(g.pg, "barrier"):
        try:
            g.pg.barrier(dist.BarrierOptions()).wait()
            return
        except Exception:
            pass
    t = torch.zeros(1)
    if g.backend == "rccl":
        t = t.cuda()
    g.pg.allreduce([t]).wait()


def send(tensor, dst_rank: int, group_name: str = "default"):
    g = _get(group_name)
    g.pg.send([tensor], dst_rank, 0).wait()


def recv(tensor, src_rank: int, group_name: str = "default"):
    g = _get(group_name)
    g.pg.recv([tensor], src_rank, 0).wait()


def send_multigpu(tensor, dst_rank, dst_gpu_index, group_name="default"):
    send(tensor, dst_rank, group_name)


def recv_multigpu(tensor, src_rank, src_gpu_index, group_name="default"):
    recv(tensor, src_rank, group_name)
