"""ClientServer: hosts remote ray_amd clients inside a cluster driver
(reference: python/ray/util/client/server/server.py — the proxy that
executes client ops against the real core worker).

Runs on the driver's event loop; every blocking operation is executed on
a thread pool with the normal sync API. Per-connection state: pinned
ObjectRefs (released when the client drops its last reference or
disconnects) and created actors (non-detached ones are killed on
disconnect).
"""
from __future__ import annotations

import asyncio
import threading
from concurrent.futures import ThreadPoolExecutor
from typing import Dict

import cloudpickle

from .._core import runtime as _rtmod
from .._core.protocol import RpcServer


class ClientServer:
    def __init__(self, host: str = "127.0.0.1", port: int = 10001):
        self.host = host
        self.port = port
        self.server = RpcServer()
        self._pool = ThreadPoolExecutor(16, thread_name_prefix="ray-client")
        # id(proto) -> {"refs": {oid: ObjectRef}, "actors": set[bytes]}
        self._sessions: Dict[int, dict] = {}
        self._lock = threading.Lock()
        # msgpack-native task registry for non-Python clients (C++):
        # name -> RemoteFunction, registered by the hosting driver
        self._tasks: Dict[str, object] = {}
        for m in ("c_init c_put c_get c_wait c_export c_task c_cancel "
                  "c_actor_create c_actor_call c_kill c_gcs "
                  "c_release c_actor_msgpack c_task_msgpack").split():
            self.server.route(m, getattr(self, "h_" + m))
        self.server.on_conn_lost = self._conn_lost

    def register_task(self, name: str, remote_fn):
        """Expose a @ray.remote function to msgpack-native (e.g. C++)
        clients under `name` (reference: C++ worker task registry,
        src/ray/internal/internal.h)."""
        self._tasks[name] = remote_fn
        return self

    def start(self):
        """Start serving (requires ray_amd.init() in this process)."""
        rt = _rtmod.global_runtime()
        self.rt = rt
        fut = asyncio.run_coroutine_threadsafe(
            self.server.start_tcp(self.host, self.port), rt.loop
        )
        self.port = fut.result(30)
        return self.port

    def stop(self):
        asyncio.run_coroutine_threadsafe(self.server.close(), self.rt.loop)

    def _session(self, conn) -> dict:
        with self._lock:
            s = self._sessions.get(id(conn))
            if s is None:
                s = self._sessions[id(conn)] = {"refs": {}, "actors": set()}
            return s

    def _conn_lost(self, proto, exc):
        s = self._sessions.pop(id(proto), None)
        if s is None:
            return

        def cleanup():
            import ray_amd as ray

            s["refs"].clear()  # drops pins -> server refcounts release
            for aid in s["actors"]:
                try:
                    ray.kill(
                        _handle(aid)
                    )
                except Exception:
                    pass

        self._pool.submit(cleanup)

    def _pin(self, conn, refs):
        s = self._session(conn)
        for r in refs:
            s["refs"][r.id] = r

    async def _off(self, fn, *a):
        return await self.rt.loop.run_in_executor(self._pool, fn, *a)

    # ---------------- handlers ----------------

    async def h_c_init(self, conn, p):
        self._session(conn)
        return {
            "session_dir": self.rt.session_dir,
            "node_id": self.rt.node_id,
            "job_id": self.rt.job_id,
        }

    async def h_c_put(self, conn, p):
        def do():
            import ray_amd as ray

            return ray.put(cloudpickle.loads(p["value"]))

        ref = await self._off(do)
        self._pin(conn, [ref])
        return {"id": ref.id, "owner": ref.owner_addr}

    async def h_c_get(self, conn, p):
        def do():
            import ray_amd as ray

            refs = [_rtmod.ObjectRef(bytes(i), o) for i, o in p["refs"]]
            try:
                vals = ray.get(refs, timeout=p.get("timeout"))
            except BaseException as e:  # noqa
                return {"error": cloudpickle.dumps(e)}
            return {"error": None,
                    "values": [cloudpickle.dumps(v) for v in vals]}

        return await self._off(do)

    async def h_c_wait(self, conn, p):
        def do():
            import ray_amd as ray

            refs = [_rtmod.ObjectRef(bytes(i), o) for i, o in p["refs"]]
            ready, rest = ray.wait(
                refs, num_returns=p["num_returns"],
                timeout=p.get("timeout"),
                fetch_local=p.get("fetch_local", True),
            )
            return {"ready": [r.id for r in ready],
                    "not_ready": [r.id for r in rest]}

        return await self._off(do)

    async def h_c_export(self, conn, p):
        await self.rt._export_function(bytes(p["fn_id"]), p["fn"])
        return True

    async def h_c_task(self, conn, p):
        def do():
            args_tuple = cloudpickle.loads(p["args"])
            refs = self.rt.submit_task(
                None, bytes(p["fn_id"]), p["name"], args_tuple,
                dict(p["options"]),
            )
            return refs

        refs = await self._off(do)
        self._pin(conn, refs)
        return {"refs": [(r.id, r.owner_addr) for r in refs]}

    async def h_c_cancel(self, conn, p):
        self.rt.cancel_task(bytes(p["id"]))
        return True

    async def h_c_actor_create(self, conn, p):
        def do():
            args_tuple = cloudpickle.loads(p["args"])
            return self.rt.create_actor(
                bytes(p["key"]), p["cls"], dict(p["options"]), args_tuple
            )

        aid = await self._off(do)
        opts = p["options"] or {}
        if opts.get("lifetime") != "detached":
            self._session(conn)["actors"].add(aid)
        return {"actor_id": aid}

    async def h_c_actor_call(self, conn, p):
        def do():
            args_tuple = cloudpickle.loads(p["args"])
            return self.rt.submit_actor_task(
                bytes(p["actor_id"]), p["method"], args_tuple,
                dict(p["options"]),
            )

        refs = await self._off(do)
        self._pin(conn, refs)
        return {"refs": [(r.id, r.owner_addr) for r in refs]}

    async def h_c_kill(self, conn, p):
        def do():
            self.rt.kill_actor(bytes(p["actor_id"]),
                               p.get("no_restart", True))

        await self._off(do)
        self._session(conn)["actors"].discard(bytes(p["actor_id"]))
        return True

    async def h_c_gcs(self, conn, p):
        return await self.rt._gcs_rpc(p["method"], p["payload"])

    # ---- msgpack-native surface (C++ / non-Python clients) ----

    async def h_c_actor_msgpack(self, conn, p):
        """Call a NAMED actor with plain-data (msgpack) args; the result
        must be plain data too. No pickle anywhere — usable from C++."""

        def do():
            import ray_amd as ray

            try:
                h = ray.get_actor(p["name"], namespace=p.get("namespace"))
                ref = getattr(h, p["method"]).remote(
                    *(p.get("args") or []), **(p.get("kwargs") or {})
                )
                v = ray.get(ref, timeout=p.get("timeout", 60.0))
                from .._core.protocol import pack

                pack(v)  # plain-data check: fail here, not in the reply
                return {"ok": True, "value": v}
            except BaseException as e:  # noqa
                return {"ok": False, "error": f"{type(e).__name__}: {e}"}

        return await self._off(do)

    async def h_c_task_msgpack(self, conn, p):
        """Run a registered task with plain-data args and return its
        plain-data result."""

        def do():
            import ray_amd as ray

            fn = self._tasks.get(p["name"])
            if fn is None:
                return {"ok": False,
                        "error": f"no registered task {p['name']!r}"}
            try:
                ref = fn.remote(*(p.get("args") or []),
                                **(p.get("kwargs") or {}))
                v = ray.get(ref, timeout=p.get("timeout", 60.0))
                from .._core.protocol import pack

                pack(v)
                return {"ok": True, "value": v}
            except BaseException as e:  # noqa
                return {"ok": False, "error": f"{type(e).__name__}: {e}"}

        return await self._off(do)

    async def h_c_release(self, conn, p):
        s = self._session(conn)
        for i in p["ids"]:
            s["refs"].pop(bytes(i), None)
        return True


def _handle(actor_id: bytes):
    from ..api import ActorHandle

    return ActorHandle(actor_id)
