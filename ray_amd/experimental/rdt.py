"""RDT — GPU-native tensor transport between actors.

Reference: python/ray/experimental/rdt/ (RDTStore rdt_store.py:159,
cuda_ipc_transport.py:35-101). This is the north star's hipIpc
zero-copy GPU object store: tensors stay in HBM; passing one between
co-located actors ships a hipIpcMemHandle (via
torch.multiprocessing.reductions.reduce_tensor, which uses hipIpc on
ROCm — dmabuf mode, HSA_ENABLE_IPC_MODE_LEGACY=0) instead of copying
through host shared memory. Cross-node or cross-GPU transfers fall back
to a staged copy (RCCL p2p is the planned fast path).

Usage inside actors:

    store = rdt.get_gpu_object_store()
    ref = store.put(gpu_tensor)          # -> GpuObjectRef (tiny, picklable)
    ...pass ref through normal ray calls...
    t = store.get(ref)                   # zero-copy view on the same node
"""
from __future__ import annotations

import os
import threading
from typing import Dict, Optional, Tuple

import cloudpickle

_store: Optional["GpuObjectStore"] = None
_lock = threading.Lock()


class _RdtIO:
    """Dedicated I/O loop for RDT fetches: store.get() may be invoked
    during DESERIALIZATION on the runtime's event-loop thread (auto-
    fetching refs in task args), where blocking on the runtime loop
    would deadlock. Fetch RPCs run on this loop instead."""

    def __init__(self):
        import asyncio

        self.loop = asyncio.new_event_loop()
        self._conns = {}
        t = threading.Thread(target=self.loop.run_forever,
                             name="ray_amd_rdt_io", daemon=True)
        t.start()

    async def _conn(self, addr):
        from ray_amd._core.protocol import RpcClient

        c = self._conns.get(addr)
        if c is None or not c.connected:
            c = RpcClient()
            await c.connect(addr, retries=10)
            self._conns[addr] = c
        return c

    def call(self, addr, method, payload, timeout=120.0):
        import asyncio

        async def _do():
            c = await self._conn(addr)
            return await c.call(method, payload)

        return asyncio.run_coroutine_threadsafe(
            _do(), self.loop).result(timeout)

    def notify(self, addr, method, payload):
        import asyncio

        async def _do():
            try:
                c = await self._conn(addr)
                c.notify(method, payload)
            except Exception:
                pass

        asyncio.run_coroutine_threadsafe(_do(), self.loop)


_io: Optional[_RdtIO] = None


def _rdt_io() -> _RdtIO:
    global _io
    with _lock:
        if _io is None:
            _io = _RdtIO()
        return _io


class GpuObjectRef:
    """Tiny handle: (object id, owner runtime address, node id, meta)."""

    __slots__ = ("id", "owner_addr", "node_id", "shape", "dtype", "device")

    def __init__(self, oid: bytes, owner_addr: str, node_id: bytes,
                 shape, dtype, device):
        self.id = oid
        self.owner_addr = owner_addr
        self.node_id = node_id
        self.shape = shape
        self.dtype = dtype
        self.device = device

    def __reduce__(self):
        return (
            GpuObjectRef,
            (self.id, self.owner_addr, self.node_id, self.shape, self.dtype,
             self.device),
        )

    def __repr__(self):
        return f"GpuObjectRef({self.id.hex()[:12]}, {self.shape}, {self.dtype})"


class GpuObjectStore:
    """Per-process store of GPU tensors exported over hipIpc."""

    def __init__(self):
        from ray_amd._core import runtime as _rt

        self._rt = _rt.global_runtime()
        self._tensors: Dict[bytes, "torch.Tensor"] = {}
        self._cache: Dict[bytes, "torch.Tensor"] = {}
        self._events: Dict[bytes, object] = {}
        # observability: did the last ipc fetch fall back to a staged
        # copy because the mapped view failed checksum verification?
        self.last_fetch_fallback = False
        self._rt.server.route("rdt_fetch", self._h_fetch)
        self._rt.server.route("rdt_free", self._h_free)

    # ---------------- owner side ----------------

    def put(self, tensor) -> GpuObjectRef:
        import torch

        assert isinstance(tensor, torch.Tensor)
        oid = os.urandom(16)
        self._tensors[oid] = tensor
        return GpuObjectRef(
            oid, self._rt.addr, self._rt.node_id, tuple(tensor.shape),
            str(tensor.dtype), str(tensor.device),
        )

    def _h_fetch(self, conn, p):
        import torch
        from torch.multiprocessing import reductions

        oid = bytes(p["id"])
        t = self._tensors.get(oid)
        if t is None:
            return None
        if p.get("mode") == "ipc" and t.is_cuda:
            # Producer-stream ordering: export an interprocess hipEvent
            # for the consumer's stream to wait on (reference:
            # cuda_ipc_transport.py:71-101) AND synchronize the
            # producing device before handing out the handle — the
            # imported-event wait was observed to miss rarely on the
            # dmabuf IPC path (consumer read uninitialized memory
            # ~1/5 runs), and a device sync at RDT-fetch frequency is
            # noise next to the RPC.
            ev_handle = None
            try:
                ev = torch.cuda.Event(interprocess=True)
                ev.record(torch.cuda.current_stream(t.device))
                ev_handle = ev.ipc_handle()
                self._events[oid] = ev  # keep alive until freed
            except Exception:
                pass
            torch.cuda.synchronize(t.device)
            func, args = reductions.reduce_tensor(t)
            flat = t.detach().view(-1)
            chk = (float(flat[:4].float().sum())
                   + float(flat[-4:].float().sum()))
            return {"mode": "ipc",
                    "payload": cloudpickle.dumps((func, args)),
                    "event": ev_handle,
                    "check": chk,
                    "device": t.device.index or 0}
        # staged fallback: device -> host -> bytes
        cpu = t.detach().cpu().contiguous()
        return {
            "mode": "staged",
            "payload": cloudpickle.dumps(
                (cpu.numpy() if cpu.dtype != torch.bfloat16
                 else cpu.view(torch.int16).numpy(), str(t.dtype))
            ),
        }

    def _h_free(self, conn, p):
        oid = bytes(p["id"])
        self._tensors.pop(oid, None)
        self._events.pop(oid, None)

    # ---------------- consumer side ----------------

    def get(self, ref: GpuObjectRef, device=None, consume: bool = False):
        import torch

        if ref.id in self._tensors:  # we are the owner
            return self._tensors[ref.id]
        if ref.id in self._cache:
            return self._cache[ref.id]
        same_node = ref.node_id == self._rt.node_id
        mode = (
            "ipc"
            if same_node
            and torch.cuda.is_available()
            and str(ref.device).startswith("cuda")
            else "staged"
        )

        r = _rdt_io().call(ref.owner_addr, "rdt_fetch",
                           {"id": ref.id, "mode": mode})
        if r is None:
            raise KeyError(f"GPU object {ref.id.hex()} not found at owner")
        if r["mode"] == "ipc":
            torch.cuda.ipc_collect()  # drop stale cached handle maps
            func, args = cloudpickle.loads(r["payload"])
            t = func(*args)
            if r.get("event") is not None:
                # order after the producer's stream without host sync
                ev = torch.cuda.Event.from_ipc_handle(
                    r.get("device", 0), r["event"]
                )
                torch.cuda.current_stream(t.device).wait_event(ev)
            if r.get("check") is not None:
                # verify the mapping (a rare dmabuf-import fault was
                # observed returning garbage): checksum the edges and
                # fall back to a staged copy on mismatch
                flat = t.detach().view(-1)
                got = (float(flat[:4].float().sum())
                       + float(flat[-4:].float().sum()))
                import math

                bad = (math.isnan(got) != math.isnan(r["check"])
                       or (not math.isnan(got)
                           and abs(got - r["check"]) > 1e-3
                           + 1e-4 * abs(r["check"])))
                self.last_fetch_fallback = bool(bad)
                if bad:
                    r2 = _rdt_io().call(ref.owner_addr, "rdt_fetch",
                                        {"id": ref.id, "mode": "staged"})
                    arr, dtype_str = cloudpickle.loads(r2["payload"])
                    t = torch.from_numpy(arr.copy())
                    if dtype_str == "torch.bfloat16":
                        t = t.view(torch.bfloat16)
                    t = t.to("cuda")
        else:
            arr, dtype_str = cloudpickle.loads(r["payload"])
            t = torch.from_numpy(arr.copy())
            if dtype_str == "torch.bfloat16":
                t = t.view(torch.bfloat16)
            if device is None and torch.cuda.is_available():
                device = "cuda"
            if device is not None:
                t = t.to(device)
        self._cache[ref.id] = t
        if consume:
            self.free(ref)
        return t

    def free(self, ref: GpuObjectRef):
        self._cache.pop(ref.id, None)
        if ref.id in self._tensors:
            del self._tensors[ref.id]
            return

        _rdt_io().notify(ref.owner_addr, "rdt_free", {"id": ref.id})

    def num_objects(self) -> int:
        return len(self._tensors)


def get_gpu_object_store() -> GpuObjectStore:
    global _store
    with _lock:
        if _store is None:
            _store = GpuObjectStore()
        return _store


# --------------------------------------------------------------------------
# tensor_transport integration (reference: actor.py:621 tensor_transport
# option + rdt_manager.py __ray_send__/__ray_recv__ injection): GPU
# tensors in actor-call args/returns (and compiled-DAG channel payloads)
# are replaced with auto-fetching refs, so same-node consumers get a
# zero-copy hipIpc view instead of a host-staged copy.
# --------------------------------------------------------------------------


def _fetch_on_load(oid, owner_addr, node_id, shape, dtype, device, consume):
    ref = GpuObjectRef(oid, owner_addr, node_id, shape, dtype, device)
    return get_gpu_object_store().get(ref, consume=consume)


class _AutoFetchRef:
    """Pickles as a thunk that fetches the tensor wherever it is
    deserialized (driver or another actor)."""

    __slots__ = ("ref", "consume")

    def __init__(self, ref: GpuObjectRef, consume: bool = False):
        self.ref = ref
        self.consume = consume

    def __reduce__(self):
        r = self.ref
        return (_fetch_on_load,
                (r.id, r.owner_addr, r.node_id, r.shape, r.dtype, r.device,
                 self.consume))


def offload_tensors(value, consume: bool = False, _refs=None):
    """Recursively replace CUDA torch tensors in value with
    auto-fetching GPU-store refs. Returns (new_value, refs)."""
    if _refs is None:
        _refs = []
    try:
        import torch
    except ImportError:
        return value, _refs
    if isinstance(value, torch.Tensor) and value.is_cuda:
        ref = get_gpu_object_store().put(value)
        _refs.append(ref)
        return _AutoFetchRef(ref, consume), _refs
    if isinstance(value, (list, tuple)):
        out = [offload_tensors(v, consume, _refs)[0] for v in value]
        return (type(value)(out) if not isinstance(value, tuple)
                else tuple(out)), _refs
    if isinstance(value, dict):
        return {k: offload_tensors(v, consume, _refs)[0]
                for k, v in value.items()}, _refs
    return value, _refs


def has_cuda_tensors(value) -> bool:
    try:
        import torch
    except ImportError:
        return False
    if isinstance(value, torch.Tensor):
        return value.is_cuda
    if isinstance(value, (list, tuple)):
        return any(has_cuda_tensors(v) for v in value)
    if isinstance(value, dict):
        return any(has_cuda_tensors(v) for v in value.values())
    return False
