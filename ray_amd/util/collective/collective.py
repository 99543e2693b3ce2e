"""Declarative collective groups over actors (reference:
python/ray/util/collective/collective.py:149-624).

MI355X-native backend: GPU groups run on ray_amd's own RCCL
communicator extension (csrc/rccl_comm.hip — rcclCommInitRank with the
unique id exchanged through the GCS KV, a dedicated high-priority HIP
stream + events per group, bf16 collectives over xGMI). This owns the
layer the reference delegates to cupy
(collective_group/nccl_collective_group.py:126); bf16 IS supported
(the reference notes cupy couldn't, nccl_util.py:693). "torch_gloo" is
the CPU path (torch ProcessGroupGloo on a TCPStore).

Rendezvous runs through the GCS KV: rank 0 publishes the RCCL unique
id (GPU) or host:port (CPU) under "collective:<group>"; everyone else
polls. Outside a ray_amd session (e.g. plain torchrun)
MASTER_ADDR/MASTER_PORT env are used to share it over a TCPStore.
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


def _load_native():
    """Import the in-tree RCCL extension; loud failure on GPU boxes —
    a silent torch-PG fallback would defeat the native-collective
    contract (driver checks which .so files actually loaded)."""
    from ... import _rccl_comm  # in-tree ray_amd/_rccl_comm.so

    return _rccl_comm


class Group:
    def __init__(self, pg, rank: int, world_size: int, backend: str, store=None):
        self.pg = pg
        self.rank = rank
        self.world_size = world_size
        self.backend = backend
        self._store = store  # keep TCPStore alive


class NativeRcclGroup:
    """A collective group on the native RCCL comm (one per group per
    process; dedicated comm stream lives inside the extension)."""

    backend = "rccl"

    def __init__(self, comm):
        self.comm = comm
        self.rank = comm.rank()
        self.world_size = comm.world_size()


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


def _kv_share_bytes(group_name: str, rank: int, payload: Optional[bytes],
                    timeout: float = 120.0) -> bytes:
    """Rank 0 publishes `payload` under the group key; other ranks poll
    it back. Used to exchange the RCCL unique id (reference:
    nccl_collective_group.py rendezvous through a named store actor)."""
    from ..._core import runtime as _rt

    key = f"collective_uid:{group_name}".encode()
    if _rt.is_initialized():
        rt = _rt.global_runtime()
        if rank == 0:
            rt.gcs_call("kv_put", {"ns": "collective", "key": key,
                                   "value": payload})
            return payload
        deadline = time.time() + timeout
        while time.time() < deadline:
            v = rt.gcs_call("kv_get", {"ns": "collective", "key": key})
            if v:
                return bytes(v)
            time.sleep(0.02)
        raise TimeoutError(f"uid rendezvous for group {group_name} timed out")
    # outside a ray session (torchrun): share over a TCPStore
    host = os.environ.get("MASTER_ADDR", "127.0.0.1")
    port = int(os.environ.get("MASTER_PORT", "29500"))
    store = dist.TCPStore(host, port, None, is_master=(rank == 0),
                          timeout=datetime.timedelta(seconds=timeout))
    skey = f"rccl_uid:{group_name}"
    if rank == 0:
        store.set(skey, payload)
        return payload
    return bytes(store.get(skey))


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

    if backend in ("rccl", "nccl") and torch.cuda.is_available() \
            and not os.environ.get("RAY_AMD_COLLECTIVE_TORCH_PG"):
        native = _load_native()
        uid = native.unique_id() if rank == 0 else None
        uid = _kv_share_bytes(group_name, rank, uid)
        comm = native.RcclComm(world_size, rank, uid,
                               torch.cuda.current_device())
        _groups[group_name] = NativeRcclGroup(comm)
        return

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


def _red_name(op: ReduceOp) -> str:
    return {ReduceOp.SUM: "sum", ReduceOp.PRODUCT: "prod",
            ReduceOp.MIN: "min", ReduceOp.MAX: "max"}.get(op, "sum")


def allreduce(tensor, group_name: str = "default", op: ReduceOp = ReduceOp.SUM):
    g = _get(group_name)
    if isinstance(g, NativeRcclGroup):
        g.comm.allreduce(tensor, _red_name(op))
        return
    opts = dist.AllreduceOptions()
    opts.reduceOp = torch_reduce_op(op)
    g.pg.allreduce([tensor], opts).wait()


def allreduce_multigpu(tensor_list, group_name="default", op=ReduceOp.SUM):
    for t in tensor_list:
        allreduce(t, group_name, op)


def reduce(tensor, dst_rank: int = 0, group_name: str = "default",
           op: ReduceOp = ReduceOp.SUM):
    g = _get(group_name)
    if isinstance(g, NativeRcclGroup):
        g.comm.reduce(tensor, dst_rank, _red_name(op))
        return
    opts = dist.ReduceOptions()
    opts.rootRank = dst_rank
    opts.reduceOp = torch_reduce_op(op)
    g.pg.reduce([tensor], opts).wait()


def broadcast(tensor, src_rank: int = 0, group_name: str = "default"):
    g = _get(group_name)
    if isinstance(g, NativeRcclGroup):
        g.comm.broadcast(tensor, src_rank)
        return
    opts = dist.BroadcastOptions()
    opts.rootRank = src_rank
    g.pg.broadcast([tensor], opts).wait()


def allgather(tensor_list: List, tensor, group_name: str = "default"):
    g = _get(group_name)
    if isinstance(g, NativeRcclGroup):
        out = torch.empty(
            (g.world_size,) + tuple(tensor.shape),
            dtype=tensor.dtype, device=tensor.device,
        )
        g.comm.allgather(out, tensor.contiguous())
        for i, t in enumerate(tensor_list):
            t.copy_(out[i].view_as(t))
        return
    g.pg.allgather([tensor_list], [tensor]).wait()


def reducescatter(tensor, tensor_list: List, group_name: str = "default",
                  op: ReduceOp = ReduceOp.SUM):
    g = _get(group_name)
    if isinstance(g, NativeRcclGroup):
        inp = torch.cat([t.contiguous().flatten() for t in tensor_list])
        out = torch.empty_like(tensor).flatten()
        g.comm.reducescatter(out, inp, _red_name(op))
        tensor.copy_(out.view_as(tensor))
        return
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
    if isinstance(g, NativeRcclGroup):
        g.comm.barrier()
        return
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
    if isinstance(g, NativeRcclGroup):
        g.comm.send(tensor, dst_rank)
        return
    g.pg.send([tensor], dst_rank, 0).wait()


def recv(tensor, src_rank: int, group_name: str = "default"):
    g = _get(group_name)
    if isinstance(g, NativeRcclGroup):
        g.comm.recv(tensor, src_rank)
        return
    g.pg.recv([tensor], src_rank, 0).wait()


def send_multigpu(tensor, dst_rank, dst_gpu_index, group_name="default"):
    send(tensor, dst_rank, group_name)


def recv_multigpu(tensor, src_rank, src_gpu_index, group_name="default"):
    recv(tensor, src_rank, group_name)
