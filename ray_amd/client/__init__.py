"""Ray-Client equivalent: drive a remote ray_amd cluster from a thin
client process (reference: python/ray/util/client/ — ray.init("ray://...")).

Architecture (re-designed, no gRPC): the client installs a
`ClientRuntime` as the process's global runtime. It implements the same
method surface `ray_amd.api` calls on a real CoreRuntime, but every
operation is one msgpack-RPC to a `ClientServer` (client/server.py)
running inside a driver on the cluster. ObjectRefs round-trip by
(id, owner_addr) — the server pins a real ref per client ref and
releases it when the client's refcount drops to zero or the connection
closes.

    # on the cluster head
    import ray_amd as ray
    from ray_amd.client.server import ClientServer
    ray.init()
    ClientServer(port=10001).start()

    # anywhere that can reach the head over TCP
    import ray_amd as ray
    ray.init("ray_amd://127.0.0.1:10001")
    ray.get(f.remote(2))
"""
from __future__ import annotations

import asyncio
import threading
from typing import Any, Dict, List, Optional

import cloudpickle

from .._core.protocol import RpcClient

_CALL_SLACK = 30.0  # extra wall time allowed beyond a user timeout


class ClientRuntime:
    """Global-runtime stand-in that proxies to a ClientServer."""

    is_client = True

    def __init__(self, addr: str, namespace: Optional[str] = None):
        # addr: "host:port"
        self.addr = addr
        self.namespace = namespace
        self._closed = False
        self.loop = asyncio.new_event_loop()
        self._client = RpcClient()
        self._refs: Dict[bytes, int] = {}
        self._refs_lock = threading.Lock()
        self._loop_thread: Optional[threading.Thread] = None
        self.session_dir = ""
        self.node_id = b""
        self.job_id = 0
        self.worker_id = b"client"
        self.raylet_addr = ""
        self.gcs_addr = ""

    # ---- lifecycle ----

    def connect(self):
        t = threading.Thread(target=self._loop_main, daemon=True,
                             name="ray_amd-client-loop")
        t.start()
        self._loop_thread = t
        hello = self._call_sync(self._connect_and_hello(), timeout=60)
        self.session_dir = hello.get("session_dir", "")
        self.node_id = bytes(hello.get("node_id", b""))
        self.job_id = hello.get("job_id", 0)

    async def _connect_and_hello(self):
        host, port = self.addr.rsplit(":", 1)
        await self._client.connect(f"tcp:{host}:{port}", retries=40)
        return await self._client.call(
            "c_init", {"namespace": self.namespace}
        )

    def _loop_main(self):
        asyncio.set_event_loop(self.loop)
        self.loop.run_forever()

    def shutdown(self):
        if self._closed:
            return
        self._closed = True
        from .._core import runtime as _rtmod

        if _rtmod._runtime is self:
            _rtmod.set_global_runtime(None)
        try:
            # let queued release notifications start before stopping
            asyncio.run_coroutine_threadsafe(
                asyncio.sleep(0.02), self.loop
            ).result(1)
        except Exception:
            pass
        try:
            self.loop.call_soon_threadsafe(self._client.close)
            self.loop.call_soon_threadsafe(self.loop.stop)
        except Exception:
            pass

    # ---- plumbing the api layer relies on ----

    def _run(self, coro):
        return asyncio.run_coroutine_threadsafe(coro, self.loop)

    def _call_sync(self, coro, timeout=None):
        return self._run(coro).result(timeout)

    async def _rpc(self, method: str, payload: Any, timeout=None):
        if timeout is not None:
            return await asyncio.wait_for(
                self._client.call(method, payload), timeout + _CALL_SLACK
            )
        return await self._client.call(method, payload)

    def _rpc_sync(self, method: str, payload: Any, timeout=None):
        if self._closed:
            raise RuntimeError("ray client connection closed")
        return self._call_sync(self._rpc(method, payload, timeout))

    # ---- reference counting (client side mirrors, server pins) ----

    def _add_local_ref(self, oid: bytes, owner_addr: str):
        with self._refs_lock:
            self._refs[oid] = self._refs.get(oid, 0) + 1

    def _remove_local_ref(self, oid: bytes):
        with self._refs_lock:
            n = self._refs.get(oid)
            if n is None:
                return
            if n > 1:
                self._refs[oid] = n - 1
                return
            del self._refs[oid]
        if not self._closed:
            try:
                self._run(self._rpc("c_release", {"ids": [oid]}))
            except Exception:
                pass

    # ---- object API ----

    def put(self, value, _owner=None):
        from .._core.runtime import ObjectRef

        r = self._rpc_sync("c_put", {"value": cloudpickle.dumps(value)})
        return ObjectRef(bytes(r["id"]), r["owner"])

    def get_sync(self, refs, timeout=None):
        r = self._rpc_sync(
            "c_get",
            {"refs": [(x.id, x.owner_addr) for x in refs],
             "timeout": timeout},
            timeout=timeout,
        )
        if r.get("error") is not None:
            raise cloudpickle.loads(r["error"])
        return [cloudpickle.loads(v) for v in r["values"]]

    def wait_sync(self, refs, num_returns=1, timeout=None, fetch_local=True):
        r = self._rpc_sync(
            "c_wait",
            {"refs": [(x.id, x.owner_addr) for x in refs],
             "num_returns": num_returns, "timeout": timeout,
             "fetch_local": fetch_local},
            timeout=timeout,
        )
        by_id = {x.id: x for x in refs}
        ready = [by_id[bytes(i)] for i in r["ready"]]
        rest = [by_id[bytes(i)] for i in r["not_ready"]]
        return ready, rest

    # ---- tasks ----

    async def _export_function(self, fn_id: bytes, pickled: bytes):
        await self._rpc("c_export", {"fn_id": fn_id, "fn": pickled})

    def submit_task(self, pickled_fn, fn_id, name, args_tuple, options):
        from .._core.runtime import ObjectRef

        if options.get("num_returns") == "streaming":
            raise NotImplementedError(
                "streaming generators are not supported over the client"
            )
        r = self._rpc_sync(
            "c_task",
            {"fn_id": fn_id, "name": name,
             "args": cloudpickle.dumps(args_tuple),
             "options": _clean_options(options)},
        )
        return [ObjectRef(bytes(i), o) for i, o in r["refs"]]

    def cancel_task(self, return_oid: bytes):
        self._rpc_sync("c_cancel", {"id": return_oid})

    # ---- actors ----

    def create_actor(self, key, pickled_cls, options, args_tuple) -> bytes:
        r = self._rpc_sync(
            "c_actor_create",
            {"key": key, "cls": pickled_cls,
             "args": cloudpickle.dumps(args_tuple),
             "options": _clean_options(options)},
        )
        return bytes(r["actor_id"])

    def submit_actor_task(self, actor_id, method, args_tuple, options):
        from .._core.runtime import ObjectRef

        if options.get("num_returns") == "streaming":
            raise NotImplementedError(
                "streaming generators are not supported over the client"
            )
        r = self._rpc_sync(
            "c_actor_call",
            {"actor_id": actor_id, "method": method,
             "args": cloudpickle.dumps(args_tuple),
             "options": _clean_options(options)},
        )
        return [ObjectRef(bytes(i), o) for i, o in r["refs"]]

    def kill_actor(self, actor_id: bytes, no_restart: bool = True):
        self._rpc_sync("c_kill", {"actor_id": actor_id,
                                  "no_restart": no_restart})

    # ---- GCS passthrough (named actors, node table, kv, ...) ----

    def gcs_call(self, method: str, payload: dict):
        return self._rpc_sync("c_gcs", {"method": method,
                                        "payload": payload})

    async def _gcs_rpc(self, method, payload):
        return await self._rpc("c_gcs", {"method": method,
                                         "payload": payload})


def _clean_options(options: dict) -> dict:
    """Options must cross the wire as msgpack — strip non-plain values
    (placement group handles were normalized to tuples already)."""
    out = {}
    for k, v in options.items():
        if v is None or isinstance(v, (bool, int, float, str, bytes)):
            out[k] = v
        elif isinstance(v, (list, tuple)):
            out[k] = list(v)
        elif isinstance(v, dict):
            out[k] = v
    return out
