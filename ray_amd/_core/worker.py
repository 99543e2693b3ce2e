"""Worker process — executes normal tasks and hosts actors.

Counterpart of the reference's worker side of CoreWorker: task receiver
queues (task_execution/task_receiver.h:43), ordered actor execution,
concurrency groups via a thread pool, async-actor coroutines on the
event loop, and the Python execution callback
(_raylet.pyx:2436 execute_task_with_cancellation_handler).

One process == one worker == (optionally) one actor, one asyncio loop on
the main thread, user code on executor thread(s) so the loop stays
responsive for serving owned objects.
"""
from __future__ import annotations

import asyncio
import concurrent.futures
import os
import sys
import threading
import time
import traceback

# Driver sys.path propagation (so by-reference pickled functions import).
_pp = os.environ.get("RAY_AMD_PYTHONPATH")
if _pp:
    import sys as _sys

    for _p in reversed(_pp.split(os.pathsep)):
        if _p and _p not in _sys.path:
            _sys.path.insert(0, _p)

# GPU pinning must happen before torch import anywhere in this process.
_gpu_ids = os.environ.get("RAY_AMD_GPU_IDS")
if _gpu_ids:
    os.environ.setdefault("CUDA_VISIBLE_DEVICES", _gpu_ids)
    os.environ.setdefault("HIP_VISIBLE_DEVICES", _gpu_ids)

from .. import exceptions as exc  # noqa: E402
from . import ids, serialization, store  # noqa: E402
from .runtime import CoreRuntime, ObjectRef, set_global_runtime  # noqa: E402


def _apply_code_env(spec: dict):
    """working_dir / py_modules (already staged into the session dir by
    the driver): chdir + sys.path injection."""
    import sys as _sys

    wd = spec.get("working_dir")
    if wd and os.path.isdir(wd):
        os.chdir(wd)
        if wd not in _sys.path:
            _sys.path.insert(0, wd)
    for m in spec.get("py_modules") or []:
        if m and os.path.isdir(m) and m not in _sys.path:
            _sys.path.insert(0, m)


def _trace_activate(spec, name):
    if not spec.get("trace_ctx") and not os.environ.get("RAY_AMD_TRACING"):
        return None
    try:
        from ray_amd.util.tracing.tracing_helper import (
            activate_remote_context,
        )

        return activate_remote_context(spec.get("trace_ctx"), name)
    except Exception:
        return None


def _trace_finish(handle):
    if handle is None:
        return
    try:
        from ray_amd.util.tracing.tracing_helper import (
            finish_remote_context,
        )

        finish_remote_context(handle)
    except Exception:
        pass


class TaskContext:
    def __init__(self):
        self.task_id = None
        self.task_name = None
        self.actor_id = None
        self.gpu_ids = []


_task_ctx = TaskContext()


def current_task_context() -> TaskContext:
    return _task_ctx


class WorkerMain:
    def __init__(self):
        self.session_dir = os.environ["RAY_AMD_SESSION_DIR"]
        self.gcs_addr = os.environ["RAY_AMD_GCS_ADDR"]
        self.raylet_addr = os.environ["RAY_AMD_RAYLET_ADDR"]
        self.node_id = bytes.fromhex(os.environ["RAY_AMD_NODE_ID"])
        self.actor_id = (
            bytes.fromhex(os.environ["RAY_AMD_ACTOR_ID"])
            if os.environ.get("RAY_AMD_ACTOR_ID")
            else None
        )
        self.rt: CoreRuntime = None
        self.actor_instance = None
        self.actor_spec = None
        self.executor = concurrent.futures.ThreadPoolExecutor(
            max_workers=1, thread_name_prefix="task_exec"
        )
        self._cancelled = set()
        self._events = []
        self._max_concurrency = 1
        # producer-side streaming backpressure: task_id -> {acked, event}
        self._stream_prod = {}
        self._order_lock = None  # created on the loop in main()

    async def main(self):
        loop = asyncio.get_running_loop()
        rt = CoreRuntime(
            "worker",
            self.session_dir,
            self.gcs_addr,
            self.raylet_addr,
            self.node_id,
            loop=loop,
        )
        self.rt = rt
        rt.set_loop_thread(threading.current_thread())
        set_global_runtime(rt)
        await rt._async_start()
        rt.server.route("push_task", self.h_push_task)
        rt.server.route("actor_call", self.h_actor_call)
        rt.server.route("cancel_task", self.h_cancel_task)
        rt.server.route("exit_worker", self.h_exit_worker)
        rt.server.route("stream_ack", self.h_stream_ack)
        rt.server.route("dump_stack", self.h_dump_stack)
        self._order_lock = asyncio.Lock()
        asyncio.ensure_future(self._event_flusher())
        r = await rt.raylet.call(
            "register_worker", {"pid": os.getpid(), "addr": rt.addr}
        )
        if not r.get("ok"):
            sys.exit(1)
        if self.actor_id is not None:
            ok = await self.init_actor()
            if not ok:
                await asyncio.sleep(0.2)
                sys.exit(1)
        # Stay alive until the raylet connection drops.
        while rt.raylet.connected:
            await asyncio.sleep(0.5)
        sys.exit(0)

    def h_dump_stack(self, conn, p):
        """All-thread stack traces (reference: `ray stack` py-spy dump
        — here served in-process, no ptrace needed)."""
        import traceback as _tb

        names = {t.ident: t.name for t in threading.enumerate()}
        stacks = {}
        for tid, fr in sys._current_frames().items():
            stacks[f"{names.get(tid, 'thread')}-{tid}"] = "".join(
                _tb.format_stack(fr))
        return {"pid": os.getpid(), "actor": bool(self.actor_id),
                "stacks": stacks}

    def h_stream_ack(self, conn, p):
        """Consumer progress for a streaming generator — wakes the
        paused producer (reference: ReportGeneratorItemReturns acks)."""
        st = self._stream_prod.get(bytes(p["task_id"]))
        if st is not None:
            st["acked"] = max(st["acked"], p["consumed"])
            st["event"].set()

    # ------------- actor init -------------

    async def init_actor(self) -> bool:
        try:
            r = await self.rt.gcs.call(
                "resolve_actor", {"actor_id": self.actor_id, "wait": False}
            )
            spec_key = r.get("spec_kv_key")
            pickled_cls = await self.rt.gcs.call(
                "kv_get", {"ns": "actorcls", "key": spec_key}
            )
            args_blob = await self.rt.gcs.call(
                "kv_get", {"ns": "actorargs", "key": self.actor_id}
            )
            import cloudpickle

            self.tensor_transport = r.get("tensor_transport")
            cls, self._max_concurrency = cloudpickle.loads(pickled_cls)
            if self._max_concurrency > 1:
                self.executor = concurrent.futures.ThreadPoolExecutor(
                    max_workers=self._max_concurrency
                )
            args, kwargs = serialization.loads(args_blob)
            args, kwargs = await self._resolve_args(args, kwargs)
            _apply_code_env(r)
            _task_ctx.actor_id = self.actor_id
            _task_ctx.gpu_ids = [
                int(x) for x in (os.environ.get("RAY_AMD_GPU_IDS") or "").split(",") if x
            ]

            def _make():
                return cls(*args, **kwargs)

            self.actor_instance = await asyncio.get_running_loop().run_in_executor(
                self.executor, _make
            )
            self.rt.raylet.notify(
                "actor_ready", {"actor_id": self.actor_id, "addr": self.rt.addr}
            )
            return True
        except Exception:
            tb = traceback.format_exc()
            try:
                self.rt.raylet.notify(
                    "actor_failed", {"actor_id": self.actor_id, "error": tb}
                )
            except Exception:
                pass
            sys.stderr.write(tb)
            return False

    # ------------- arg resolution -------------

    async def _resolve_args(self, args, kwargs):
        """Replace top-level ObjectRefs with their values (reference
        semantics: only top-level args are resolved)."""
        refs = [a for a in args if isinstance(a, ObjectRef)]
        refs += [v for v in kwargs.values() if isinstance(v, ObjectRef)]
        if refs:
            vals = await self.rt.get_async(refs, timeout=600.0)
            table = dict(zip([r.id for r in refs], vals))
            args = tuple(
                table[a.id] if isinstance(a, ObjectRef) else a for a in args
            )
            kwargs = {
                k: table[v.id] if isinstance(v, ObjectRef) else v
                for k, v in kwargs.items()
            }
        return args, kwargs

    async def _load_args(self, spec):
        if spec.get("args") is not None:
            payload = serialization.loads(spec["args"])
        else:
            aid, owner, node_addr = spec["args_store"]
            ref = ObjectRef(bytes(aid), owner)
            self.rt.memory_store.setdefault(
                ref.id, ("store", node_addr, 0)
            )
            # size unknown here; _materialize reads the file after wait
            ent = self.rt.memory_store[ref.id]
            payload = await self.rt._materialize(ref.id, ent)
        return payload

    # ------------- normal tasks -------------

    async def h_push_task(self, conn, spec):
        t0 = time.time()
        task_id = bytes(spec["task_id"])
        if task_id in self._cancelled:
            self._cancelled.discard(task_id)
            err = serialization.dumps(exc.TaskCancelledError(spec.get("name", "")))
            return {"status": "error", "error": err}
        try:
            fn = await self.rt.load_function(bytes(spec["fn_id"]))
            args, kwargs = await self._load_args(spec)
            args, kwargs = await self._resolve_args(args, kwargs)
        except Exception:
            return self._error_reply(spec, traceback.format_exc())
        if spec.get("env_vars"):
            os.environ.update({str(k): str(v) for k, v in spec["env_vars"].items()})
        _apply_code_env(spec)
        lease_gpus = [int(g) for g in (spec.get("gpu_ids") or [])]
        if lease_gpus and "RAY_AMD_GPU_IDS" not in os.environ:
            # Like the reference (worker.py set_gpu_ids), device
            # visibility is set per-lease in a generic pooled worker;
            # effective only if HIP has not initialized here yet.
            ids_str = ",".join(map(str, lease_gpus))
            os.environ["RAY_AMD_GPU_IDS"] = ids_str
            os.environ.setdefault("CUDA_VISIBLE_DEVICES", ids_str)
            os.environ.setdefault("HIP_VISIBLE_DEVICES", ids_str)

        loop = asyncio.get_running_loop()
        _tr = _trace_activate(spec, "task:" + (spec.get("name") or "fn"))

        def _exec():
            _task_ctx.task_id = task_id
            _task_ctx.task_name = spec.get("name")
            _task_ctx.gpu_ids = lease_gpus
            try:
                return True, fn(*args, **kwargs)
            except BaseException as e:  # noqa
                return False, e
            finally:
                _task_ctx.task_id = None

        if asyncio.iscoroutinefunction(fn):
            try:
                result = await fn(*args, **kwargs)
                ok = True
            except BaseException as e:  # noqa
                ok, result = False, e
        else:
            ok, result = await loop.run_in_executor(self.executor, _exec)
        _trace_finish(_tr)
        if ok and spec.get("streaming"):
            reply = await self._run_streaming(conn, spec, result)
        else:
            reply = self._build_reply(spec, ok, result)
        self._record_event(spec, t0, ok)
        return reply

    async def _run_streaming(self, conn, spec, gen):
        """Drive a user generator; push each yielded value to the
        caller as its own object (reference: streaming generators,
        ReportGeneratorItemReturns + generator_waiter.h backpressure)."""
        import struct

        loop = asyncio.get_running_loop()
        task_id = bytes(spec["task_id"])
        idx = 0

        def pull():
            try:
                return True, next(gen), None
            except StopIteration:
                return False, None, None
            except BaseException as e:  # noqa
                return False, None, e

        async def apull():
            try:
                return True, await gen.__anext__(), None
            except StopAsyncIteration:
                return False, None, None
            except BaseException as e:  # noqa
                return False, None, e

        is_async = hasattr(gen, "__anext__")
        if not (is_async or hasattr(gen, "__next__")):
            return self._error_reply(
                spec, "num_returns='streaming' requires a generator"
            )
        while True:
            if is_async:
                more, item, err = await apull()
            else:
                more, item, err = await loop.run_in_executor(self.executor, pull)
            if err is not None:
                name = spec.get("name", "")
                import traceback as tb

                data = serialization.dumps(
                    exc.RayTaskError(
                        name,
                        "".join(tb.format_exception(type(err), err,
                                                    err.__traceback__)),
                        None,
                    )
                )
                return {"status": "error", "error": data}
            if not more:
                break
            # Stream-item ids live in the same namespace as normal object
            # ids (nonce8 + LE u64 seq). Tag the top bit of the final byte
            # so item ids can never collide with a minted object id (a
            # normal id would need obj_seq >= 2^63): task_id is
            # nonce8+low4(task_seq), so without the tag item 0 of task N is
            # byte-identical to the object with obj_seq == task_seq(N).
            oid = task_id + struct.pack("<I", 0x80000000 | idx)
            meta, buffers, _ = self.rt._serialize_capture(item)
            size = serialization.serialized_size(meta, buffers)
            payload = {"task_id": task_id, "oid": oid}
            if size <= serialization.INLINE_MAX:
                blob = bytearray(size)
                n = serialization.write_to(memoryview(blob), meta, buffers)
                payload.update({"kind": "inline", "data": bytes(blob[:n])})
            else:
                store.put_serialized(self.rt.shm_dir, oid, meta, buffers)
                self.rt.raylet.notify("seal_object", {"id": oid, "size": size})
                payload.update(
                    {"kind": "store", "node_addr": self.raylet_addr,
                     "size": size}
                )
            payload["addr"] = self.rt.addr  # ack channel back to us
            from .protocol import MSG_NOTIFY

            conn.send([MSG_NOTIFY, 0, "stream_item", payload])
            idx += 1
            # producer backpressure (reference: generator_waiter.h:75):
            # pause while too many items sit unconsumed
            from .._config import config as _cfg

            cap = _cfg.streaming_backpressure_items
            if cap > 0:
                st = self._stream_prod.setdefault(
                    task_id, {"acked": 0, "event": asyncio.Event()}
                )
                waited = 0.0
                while idx - st["acked"] >= cap and waited < 600.0:
                    st["event"].clear()
                    try:
                        await asyncio.wait_for(st["event"].wait(), 1.0)
                    except asyncio.TimeoutError:
                        waited += 1.0
        self._stream_prod.pop(task_id, None)
        return {"status": "ok", "streaming_done": idx}

    # ------------- actor calls -------------

    async def h_actor_call(self, conn, spec):
        t0 = time.time()
        method_name = spec["method"]
        if self.actor_instance is None:
            err = serialization.dumps(
                exc.ActorDiedError("actor instance not initialized")
            )
            return {"status": "error", "error": err}
        if method_name == "__ray_terminate__":
            asyncio.get_running_loop().call_later(0.05, self._graceful_exit)
            return {"status": "ok", "results": [
                {"kind": "inline", "data": serialization.dumps(None)}]}
        # Execution-ORDER guarantee (reference: actor tasks run in
        # submission order): arg resolution awaits inside a FIFO lock so
        # a later call whose args are ready first cannot jump the queue;
        # the actual execution is awaited OUTSIDE the lock (the
        # single-thread executor preserves the dispatch order for sync
        # methods; async methods interleave by design).
        loop = asyncio.get_running_loop()
        _tr_a = _trace_activate(spec, "actor:" + method_name)
        async with self._order_lock:
            try:
                args, kwargs = await self._load_args(spec)
                args, kwargs = await self._resolve_args(args, kwargs)
                if method_name == "__ray_apply__":
                    # generic in-actor execution (reference: __ray_call__):
                    # first arg is fn(instance, *rest) — used by compiled
                    # DAG loops and debugging helpers
                    fn, args = args[0], args[1:]
                    import functools

                    method = functools.partial(fn, self.actor_instance)
                else:
                    method = getattr(self.actor_instance, method_name)
            except Exception:
                return self._error_reply(spec, traceback.format_exc())

            if asyncio.iscoroutinefunction(method):
                pending = asyncio.ensure_future(method(*args, **kwargs))
                is_async = True
            else:

                def _exec():
                    try:
                        return True, method(*args, **kwargs), False
                    except SystemExit:
                        return True, None, True
                    except BaseException as e:  # noqa
                        return False, e, False

                pending = loop.run_in_executor(self.executor, _exec)
                is_async = False

        exit_after = False
        if is_async:
            try:
                result = await pending
                ok = True
            except SystemExit:
                ok, result, exit_after = True, None, True
            except BaseException as e:  # noqa
                ok, result = False, e
        else:
            ok, result, exit_after = await pending
        _trace_finish(_tr_a)
        if ok and spec.get("streaming"):
            reply = await self._run_streaming(conn, spec, result)
        else:
            reply = self._build_reply(spec, ok, result)
        self._record_event(spec, t0, ok, method_name)
        if exit_after or method_name == "__ray_terminate__":
            loop.call_later(0.05, self._graceful_exit)
        return reply

    def _graceful_exit(self):
        try:
            self.rt.gcs.notify(
                "actor_exit",
                {"actor_id": self.actor_id, "expected": True, "cause": "exit_actor"},
            )
        except Exception:
            pass
        os._exit(0)

    def h_cancel_task(self, conn, p):
        self._cancelled.add(bytes(p["task_id"]))

    def h_exit_worker(self, conn, p):
        os._exit(0)

    # ------------- replies -------------

    def _error_reply(self, spec, tb: str, cause=None):
        name = spec.get("name") or spec.get("method", "")
        e = exc.RayTaskError(name, tb, None)
        try:
            data = serialization.dumps(e)
        except Exception:
            data = serialization.dumps(exc.RayTaskError(name, tb, None))
        return {"status": "error", "error": data}

    def _build_reply(self, spec, ok: bool, result):
        name = spec.get("name") or spec.get("method", "")
        if not ok:
            tb = "".join(
                traceback.format_exception(type(result), result, result.__traceback__)
            )
            try:
                e = exc.RayTaskError(name, tb, result)
                data = serialization.dumps(e)
            except Exception:
                data = serialization.dumps(exc.RayTaskError(name, tb, None))
            return {"status": "error", "error": data}
        num_returns = spec.get("num_returns", 1)
        if num_returns == 1:
            values = [result]
        else:
            if not isinstance(result, (tuple, list)) or len(result) != num_returns:
                return self._error_reply(
                    spec,
                    f"task declared num_returns={num_returns} but returned "
                    f"{type(result).__name__}",
                )
            values = list(result)
        results = []
        if getattr(self, "tensor_transport", None):
            # tensor_transport actors (reference: actor.py:621): GPU
            # tensors in returns ship as auto-fetching GPU-store refs
            # (zero-copy hipIpc same-node) instead of host-staged bytes
            from ray_amd.experimental import rdt as _rdt

            values = [
                _rdt.offload_tensors(v)[0] if _rdt.has_cuda_tensors(v) else v
                for v in values
            ]
        for oid, v in zip(spec["returns"], values):
            try:
                meta, buffers, _ = self.rt._serialize_capture(v)
            except Exception:
                return self._error_reply(spec, traceback.format_exc())
            size = serialization.serialized_size(meta, buffers)
            if size <= serialization.INLINE_MAX:
                blob = bytearray(size)
                n = serialization.write_to(memoryview(blob), meta, buffers)
                results.append({"kind": "inline", "data": bytes(blob[:n])})
            else:
                store.put_serialized(self.rt.shm_dir, bytes(oid), meta, buffers)
                # seal through raylet (async fire-and-forget then confirm)
                self.rt.raylet.notify("seal_object", {"id": bytes(oid), "size": size})
                results.append(
                    {"kind": "store", "node_addr": self.raylet_addr, "size": size}
                )
        return {"status": "ok", "results": results,
                "borrows": self._held_borrows(), "worker_addr": self.rt.addr}

    def _held_borrows(self):
        """Borrowed refs this worker still holds at reply time; the
        caller merges these into the owner's borrower set SYNCHRONOUSLY
        before it may drop its submitted-ref and free (closes the
        fire-and-forget borrow_add race; reference: the borrower-chain
        merge carried on task replies, reference_counter.h:44)."""
        out = []
        with self.rt._refs_lock:
            for oid, e in self.rt._refs.items():
                if e[0] > 0 and e[2] and e[2] != self.rt.addr:
                    out.append((oid, e[2]))
        return out

    async def _event_flusher(self):
        # small batches flush on a timer so timeline/state views see
        # short-lived tasks too (the 100-event flush alone starves them)
        while True:
            await asyncio.sleep(1.0)
            if self._events:
                try:
                    self.rt.raylet.notify(
                        "report_task_events", {"events": self._events}
                    )
                    self._events = []
                except Exception:
                    pass

    def _record_event(self, spec, t0, ok, method=None):
        self._events.append(
            {
                "task_id": bytes(spec["task_id"]).hex(),
                "name": method or spec.get("name", ""),
                "pid": os.getpid(),
                "start": t0,
                "end": time.time(),
                "ok": ok,
            }
        )
        if len(self._events) >= 100:
            try:
                self.rt.raylet.notify(
                    "report_task_events", {"events": self._events}
                )
                self._events = []
            except Exception:
                self._events = []


def main():
    wm = WorkerMain()
    try:
        asyncio.run(wm.main())
    except KeyboardInterrupt:
        pass


if __name__ == "__main__":
    main()
