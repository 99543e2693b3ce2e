"""Raylet — the per-node daemon.

Feature counterpart of the reference NodeManager
(src/ray/raylet/node_manager.h:146) + local lease manager
(raylet/scheduling/cluster_lease_manager.h:41) + WorkerPool
(raylet/worker_pool.h:284) + object-manager pull/push
(object_manager/object_manager.h:137) + placement-group bundle 2PC
participant (node_manager.h:621). Hosts the node's shm object store
table and serves chunked remote pulls.
"""
from __future__ import annotations

import asyncio
import os
import subprocess
import sys
import time
import traceback
from collections import deque
from typing import Dict, List, Optional

from . import ids, store
from .protocol import RpcClient, RpcServer
from .._config import config as _cfg

CHUNK = _cfg.pull_chunk_bytes  # remote pull chunk (ray_config_def.h:421)


class WorkerProc:
    def __init__(self, proc: subprocess.Popen):
        self.proc = proc
        self.addr: Optional[str] = None
        self.proto = None
        self.pid = proc.pid
        self.kind = "task"
        self.actor_id: Optional[bytes] = None
        self.idle = True
        self.lease_id: Optional[int] = None
        self.gpu_ids: List[int] = []
        self.gpu_alloc: List[tuple] = []  # [(device, fraction)]


class Lease:
    def __init__(self, lease_id, resources, worker, gpu_alloc, pg=None,
                 conn=None):
        self.lease_id = lease_id
        self.resources = resources
        self.worker: WorkerProc = worker
        self.gpu_alloc = gpu_alloc  # [(device, fraction)]
        self.gpu_ids = [d for d, _ in gpu_alloc]
        self.pg = pg
        self.conn = conn  # granting client's connection (for revocation)
        self.revoke_asked = 0.0


class Raylet:
    def __init__(
        self,
        session_dir: str,
        gcs_addr: str,
        resources: Dict[str, float],
        node_name: str = "",
        labels: Optional[dict] = None,
        object_store_memory: Optional[int] = None,
    ):
        self.session_dir = session_dir
        self.gcs_addr = gcs_addr
        self.node_id = ids.new_node_id()
        self.node_name = node_name or self.node_id.hex()[:8]
        self.resources_total = dict(resources)
        self.avail = dict(resources)
        self.labels = labels or {}
        self.server = RpcServer()
        self.gcs = RpcClient()
        self.addr: Optional[str] = None

        shm_dir = os.environ.get("RAY_AMD_SHM_DIR") or os.path.join(session_dir, "shm")
        spill_dir = os.path.join(session_dir, "spill")
        cap = object_store_memory or _cfg.object_store_memory
        self.store = store.LocalObjectStore(shm_dir, spill_dir, cap)

        self.workers: Dict[int, WorkerProc] = {}  # pid -> worker
        self._idle_task_workers: deque = deque()
        self._starting = 0
        self._lease_seq = 0
        self.leases: Dict[int, Lease] = {}
        # fair dispatch across scheduling classes (reference: fairness
        # by scheduling class in the local lease manager): one FIFO per
        # resource-shape class, round-robin between classes — a large
        # request at one class's head no longer blocks other classes
        self._pending: Dict[tuple, deque] = {}
        self._class_order: deque = deque()
        # cluster-infeasible requests waiting for new capacity
        self._infeasible: List[tuple] = []
        self._infeasible_running = False
        # pull manager (dedup + priority) and push-side flow control
        self._pulls: Dict[bytes, dict] = {}
        self._pull_heap: list = []
        self._pull_seq = 0
        self._pulls_active = 0
        self._pushes: Dict[tuple, dict] = {}
        self._worker_ready: Dict[int, asyncio.Future] = {}  # pid -> fut
        self._actor_start_futs: Dict[bytes, asyncio.Future] = {}
        self._proto_worker: Dict[int, WorkerProc] = {}
        # GPU instance pool with fractional occupancy per device
        # (reference: NodeResourceInstanceSet per-instance GPU accounting,
        # common/scheduling/resource_instance_set.h). free fraction in
        # [0,1] per device id; num_gpus=0.5 shares one device.
        ngpus = int(resources.get("GPU", 0))
        self._gpu_free: Dict[int, float] = {i: 1.0 for i in range(ngpus)}
        # reader pins: oid -> set(conn ids); per-conn reverse index so a
        # dead worker's pins are swept on disconnect
        self._pins: Dict[bytes, set] = {}
        self._conn_pins: Dict[int, set] = {}
        self._deferred_free: set = set()
        # placement-group bundles: (pg_id, idx) -> {"resources", "avail", "committed"}
        self.bundles: Dict[tuple, dict] = {}
        self._node_cache: List[dict] = []
        self._node_cache_time = 0.0

        for m in (
            "register_worker request_lease return_lease seal_object wait_object set_resource "
            "free_objects pull_object fetch_chunk object_stats start_actor "
            "actor_ready actor_failed kill_worker reserve_bundle commit_bundle "
            "rollback_bundle remove_bundle node_info ping prestart_workers "
            "report_task_events pin_object unpin_object try_recycle "
            "request_push push_ack "
            "debug_state list_workers"
        ).split():
            self.server.route(m, getattr(self, "h_" + m))
        self.server.on_conn_lost = self._conn_lost
        self._task_events: List[dict] = []

    # ---------------- lifecycle ----------------

    async def start(self):
        from .protocol import bind_server

        self.addr = await bind_server(
            self.server, self.session_dir,
            f"raylet_{self.node_id.hex()[:8]}",
        )
        await self.gcs.connect(self.gcs_addr)
        await self.gcs.call(
            "register_node",
            {
                "node_id": self.node_id,
                "addr": self.addr,
                "resources": self.resources_total,
                "labels": self.labels,
            },
        )
        n_prestart = int(min(self.resources_total.get("CPU", 0),
                            _cfg.worker_prestart))
        for _ in range(n_prestart):
            self._spawn_worker()
        asyncio.ensure_future(self._resource_reporter())

    async def _resource_reporter(self):
        # Send on change AND at least once per second: the GCS deducts
        # optimistically when placing actors, and a changed-only report
        # can race that deduction and leave the GCS cache stale forever
        # (reference: ray_syncer periodic RESOURCE_VIEW broadcasts).
        last = None
        last_time = 0.0
        while True:
            snap = dict(self.avail)
            now = time.time()
            pending = self._pending_total()
            shapes = [
                [dict(key[0]), len(q)]
                for key, q in self._pending.items() if q
            ]
            for req, fut in self._infeasible:
                if not fut.done():
                    shapes.append([dict(req), 1])
            if snap != last or now - last_time > 1.0:
                try:
                    self.gcs.notify(
                        "report_resources",
                        {"node_id": self.node_id, "available": snap,
                         "pending": pending, "pending_shapes": shapes},
                    )
                    last = snap
                    last_time = now
                except Exception:
                    pass
            await asyncio.sleep(_cfg.resource_report_period_s)

    # ---------------- worker pool ----------------

    def _spawn_worker(self, actor_spec: Optional[dict] = None) -> WorkerProc:
        env = dict(os.environ)
        # reference parity (worker defaults): cap BLAS/torch threads so
        # many workers on one box don't thrash (Ray sets this too);
        # user runtime_env env_vars can override
        env.setdefault("OMP_NUM_THREADS", "1")
        env["PYTHONUNBUFFERED"] = "1"  # live worker logs
        env["RAY_AMD_SESSION_DIR"] = self.session_dir
        env["RAY_AMD_GCS_ADDR"] = self.gcs_addr
        env["RAY_AMD_RAYLET_ADDR"] = self.addr
        env["RAY_AMD_NODE_ID"] = self.node_id.hex()
        args = [sys.executable, "-m", "ray_amd._core.worker"]
        if actor_spec is not None and actor_spec.get("profiler"):
            # profiler runtime-env plugin: launch the worker UNDER the
            # profiler (reference: runtime_env/rocprof_sys.py:17)
            args = [str(x) for x in actor_spec["profiler"]] + args
        if actor_spec is not None:
            env["RAY_AMD_ACTOR_ID"] = actor_spec["actor_id"].hex()
            for k, v in (actor_spec.get("env_vars") or {}).items():
                env[str(k)] = str(v)
        logdir = os.path.join(self.session_dir, "logs")
        os.makedirs(logdir, exist_ok=True)
        tag = (
            f"actor_{actor_spec['actor_id'].hex()[:8]}"
            if actor_spec
            else f"worker_{len(self.workers)}_{os.urandom(2).hex()}"
        )
        out = open(os.path.join(logdir, tag + ".log"), "ab", buffering=0)
        proc = subprocess.Popen(
            args, env=env, stdout=out, stderr=subprocess.STDOUT,
            start_new_session=True,
        )
        w = WorkerProc(proc)
        if actor_spec is not None:
            w.kind = "actor"
            w.actor_id = actor_spec["actor_id"]
            w.idle = False
        else:
            self._starting += 1
        self.workers[w.pid] = w
        return w

    def h_register_worker(self, conn, p):
        w = self.workers.get(p["pid"])
        if w is None:
            # worker we didn't spawn (shouldn't happen)
            return {"ok": False}
        w.addr = p["addr"]
        w.proto = conn
        self._proto_worker[id(conn)] = w
        if w.kind == "task":
            self._starting -= 1
            self._idle_task_workers.append(w)
            self._try_grant()
        fut = self._worker_ready.pop(w.pid, None)
        if fut is not None and not fut.done():
            fut.set_result(w)
        return {"ok": True, "node_id": self.node_id}

    def _conn_lost(self, proto, exc_):
        for oid in list(self._conn_pins.get(id(proto), ())):
            self._unpin(oid, id(proto))
        self._conn_pins.pop(id(proto), None)
        w = self._proto_worker.pop(id(proto), None)
        if w is None:
            return
        self.workers.pop(w.pid, None)
        if w in self._idle_task_workers:
            try:
                self._idle_task_workers.remove(w)
            except ValueError:
                pass
        if w.lease_id is not None:
            lease = self.leases.pop(w.lease_id, None)
            if lease is not None:
                self._release_resources(lease)
        if w.kind == "actor" and w.actor_id is not None:
            spec_res = getattr(w, "actor_resources", None) or {}
            bundle_key = getattr(w, "actor_bundle", None)
            self._gpu_release(w.gpu_alloc)
            if bundle_key is not None:
                b = self.bundles.get(bundle_key)
                if b is not None:
                    for k, v in spec_res.items():
                        b["avail"][k] = b["avail"].get(k, 0) + v
                # else: the bundle was already rolled back — its
                # resources went back to the node then; adding here
                # would double-release (observed: avail > total)
            else:
                for k, v in spec_res.items():
                    self.avail[k] = min(
                        self.avail.get(k, 0) + v,
                        self.resources_total.get(k, v),
                    )
            rc = w.proc.poll()
            asyncio.ensure_future(self._notify_actor_exit(w, rc))
        self._try_grant()

    async def _notify_actor_exit(self, w: WorkerProc, returncode):
        try:
            await self.gcs.call(
                "actor_exit",
                {
                    "actor_id": w.actor_id,
                    "expected": returncode == 0,
                    "cause": f"actor process exited with code {returncode}",
                },
            )
        except Exception:
            pass

    def h_prestart_workers(self, conn, p):
        for _ in range(int(p.get("n", 1))):
            self._spawn_worker()
        return True

    # ---------------- leases ----------------

    def _fits(self, avail: Dict[str, float], req: Dict[str, float]) -> bool:
        return all(avail.get(k, 0.0) + 1e-9 >= v for k, v in req.items() if v > 0)

    def _feasible_total(self, req: Dict[str, float]) -> bool:
        return all(
            self.resources_total.get(k, 0.0) + 1e-9 >= v for k, v in req.items() if v > 0
        )

    async def h_request_lease(self, conn, p):
        req = dict(p.get("resources") or {})
        pg = None
        if p.get("pg_id") is not None:
            pg = (bytes(p["pg_id"]), p.get("bundle_index"))
            bkey = self._bundle_key(pg)
            if bkey is None:
                return {"error": "placement group bundle not found"}
        if pg is None and not self._feasible_total(req):
            if p.get("no_spill"):
                # hard node-affinity / label pin: never redirect
                return {"error":
                        f"infeasible resource request {req} on pinned node"}
            spill = await self._find_spill_target(req)
            if spill:
                return {"spill": spill}
            # Cluster-infeasible RIGHT NOW: park the request and keep
            # re-evaluating — an autoscaler may add a node that fits
            # (reference: the infeasible lease queue in
            # cluster_lease_manager; demand is advertised to the
            # autoscaler through pending_shapes).
            fut = asyncio.get_running_loop().create_future()
            self._infeasible.append((req, fut))
            self._ensure_infeasible_loop()
            return await fut
        fut = asyncio.get_running_loop().create_future()
        key = (tuple(sorted(req.items())), pg)
        q = self._pending.get(key)
        if q is None:
            q = self._pending[key] = deque()
            self._class_order.append(key)
        q.append((req, pg, fut, conn))
        self._try_grant()
        return await fut

    def _pending_total(self) -> int:
        return sum(len(q) for q in self._pending.values()) + len(
            self._infeasible)

    def _ensure_infeasible_loop(self):
        if self._infeasible_running:
            return
        self._infeasible_running = True
        asyncio.ensure_future(self._infeasible_loop())

    async def _infeasible_loop(self):
        try:
            while self._infeasible:
                await asyncio.sleep(0.25)
                still = []
                for req, fut in self._infeasible:
                    if fut.done():
                        continue
                    if self._feasible_total(req):
                        # this node can now host it: re-enter the
                        # normal lease path
                        key = (tuple(sorted(req.items())), None)
                        q = self._pending.get(key)
                        if q is None:
                            q = self._pending[key] = deque()
                            self._class_order.append(key)
                        q.append((req, None, fut, None))
                        self._try_grant()
                        continue
                    spill = await self._find_spill_target(req)
                    if spill:
                        fut.set_result({"spill": spill})
                        continue
                    still.append((req, fut))
                self._infeasible = still
        finally:
            self._infeasible_running = False

    def _bundle_key(self, pg):
        pg_id, idx = pg
        if idx is not None:
            key = (pg_id, idx)
            return key if key in self.bundles else None
        # any bundle of this pg on this node
        for key in self.bundles:
            if key[0] == pg_id:
                return key
        return None

    # ---------------- GPU instance pool ----------------

    _GPU_EPS = 1e-6

    def _gpu_alloc(self, ngpu: float) -> Optional[List[tuple]]:
        """Assign device ids for a GPU request. Integer requests take
        whole free devices; a fractional request (0 < ngpu < 1) shares
        one device, best-fit packed onto the fullest device that still
        has room. Returns [(device, fraction)] or None if the request
        cannot be satisfied right now (caller must queue, never
        silently under-assign)."""
        if ngpu <= 0:
            return []
        eps = self._GPU_EPS
        if ngpu < 1.0 - eps:
            fit = [d for d, f in self._gpu_free.items() if f >= ngpu - eps]
            if not fit:
                return None
            # pack partially-used devices first to keep whole GPUs free
            dev = min(fit, key=lambda d: self._gpu_free[d])
            self._gpu_free[dev] -= ngpu
            return [(dev, ngpu)]
        n = int(round(ngpu))
        whole = [d for d, f in self._gpu_free.items() if f >= 1.0 - eps]
        if len(whole) < n:
            return None
        out = []
        for d in whole[:n]:
            self._gpu_free[d] = 0.0
            out.append((d, 1.0))
        return out

    def _gpu_release(self, alloc: List[tuple]):
        for d, frac in alloc:
            self._gpu_free[d] = min(1.0, self._gpu_free.get(d, 0.0) + frac)

    def _try_grant(self):
        """Round-robin across scheduling classes; within a class, FIFO."""
        progress = True
        while progress and self._class_order:
            progress = False
            for _ in range(len(self._class_order)):
                if not self._class_order:
                    break
                key = self._class_order[0]
                self._class_order.rotate(-1)
                q = self._pending.get(key)
                if not q:
                    self._pending.pop(key, None)
                    try:
                        self._class_order.remove(key)
                    except ValueError:
                        pass
                    continue
                if self._grant_head(q):
                    progress = True

    def _grant_head(self, q: deque) -> bool:
        """Try to grant the head of one class queue. Returns True if an
        item was consumed (granted or dropped), False if blocked."""
        req, pg, fut, rconn = q[0]
        if fut.done():
            q.popleft()
            return True
        pool_avail = self.avail
        bundle = None
        if pg is not None:
            bkey = self._bundle_key(pg)
            if bkey is None:
                q.popleft()
                fut.set_result({"error": "placement group bundle lost"})
                return True
            bundle = self.bundles[bkey]
            pool_avail = bundle["avail"]
        if not self._fits(pool_avail, req):
            if pg is None:
                self._ask_revocations(req, rconn)
            return False
        if not self._idle_task_workers:
            # bounded pool (reference: worker_pool.h soft limit):
            # never exceed what the CPU resource could run anyway
            n_task_workers = sum(
                1 for w in self.workers.values() if w.kind == "task"
            )
            cap = int(max(self.resources_total.get("CPU", 1)
                          * _cfg.worker_cap_factor, 8))
            if (
                self._starting < min(self._pending_total(), 4)
                and n_task_workers + self._starting < cap
            ):
                self._spawn_worker()
            return False
        w = self._idle_task_workers.popleft()
        if w.proc.poll() is not None or w.proto is None:
            return True
        ngpu = float(req.get("GPU", 0))
        gpu_alloc = self._gpu_alloc(ngpu)
        if gpu_alloc is None:
            # resource accounting says it fits but no device has the
            # fraction free (fragmentation) — keep the request queued
            self._idle_task_workers.appendleft(w)
            return False
        q.popleft()
        if bundle is None:
            for k, v in req.items():
                self.avail[k] = self.avail.get(k, 0) - v
        else:
            for k, v in req.items():
                bundle["avail"][k] = bundle["avail"].get(k, 0) - v
        self._lease_seq += 1
        lease = Lease(self._lease_seq, req, w, gpu_alloc, pg, conn=rconn)
        self.leases[lease.lease_id] = lease
        w.idle = False
        w.lease_id = lease.lease_id
        w.gpu_ids = lease.gpu_ids
        w.gpu_alloc = gpu_alloc
        fut.set_result(
            {"addr": w.addr, "lease_id": lease.lease_id,
             "gpu_ids": lease.gpu_ids, "raylet": self.addr}
        )
        return True

    def _ask_revocations(self, req: Dict[str, float], requester_conn):
        """Queued lease blocked on resources held by granted leases:
        ask OTHER holders to give back leases they are only caching
        idle (reference: the lease holder returns workers on
        ReleaseUnusedWorkers / idle-timeout; revocation makes that
        demand-driven instead of timer-driven). Busy leases are left
        alone — the holder returns them at task completion."""
        import time as _t

        from .protocol import MSG_NOTIFY

        now = _t.monotonic()
        for lease in self.leases.values():
            c = lease.conn
            if c is None or c is requester_conn:
                continue
            if now - lease.revoke_asked < 0.5:
                continue
            if not any(k in lease.resources for k in req):
                continue  # disjoint resources — returning it won't help
            lease.revoke_asked = now
            try:
                c.send([MSG_NOTIFY, 0, "revoke_lease",
                        {"lease_id": lease.lease_id}])
            except Exception:
                pass

    def _release_resources(self, lease: Lease):
        if lease.pg is not None:
            bkey = self._bundle_key(lease.pg)
            if bkey is not None:
                b = self.bundles[bkey]
                for k, v in lease.resources.items():
                    b["avail"][k] = b["avail"].get(k, 0) + v
        else:
            for k, v in lease.resources.items():
                self.avail[k] = self.avail.get(k, 0) + v
        self._gpu_release(lease.gpu_alloc)

    def h_return_lease(self, conn, p):
        lease = self.leases.pop(p["lease_id"], None)
        if lease is None:
            return
        self._release_resources(lease)
        w = lease.worker
        if not p.get("dead") and w.pid in self.workers and w.proc.poll() is None:
            w.idle = True
            w.lease_id = None
            self._idle_task_workers.append(w)
        self._try_grant()

    async def _find_spill_target(self, req) -> Optional[str]:
        now = time.time()
        if now - self._node_cache_time > 0.5:
            try:
                self._node_cache = await self.gcs.call("node_table", {})
                self._node_cache_time = now
            except Exception:
                return None
        feasible = []
        for n in self._node_cache:
            if not n["alive"] or n["addr"] == self.addr:
                continue
            tot = n["resources_total"]
            if all(tot.get(k, 0) + 1e-9 >= v for k, v in req.items() if v > 0):
                feasible.append(n["addr"])
        if not feasible:
            return None
        import random as _random

        return _random.choice(feasible)  # spread spillback, avoid herding

    # ---------------- actors ----------------

    async def h_start_actor(self, conn, p):
        spec = p["spec"]
        actor_id = p["actor_id"]
        spec = dict(spec)
        spec["actor_id"] = actor_id
        res = spec.get("resources") or {}
        bundle_key = None
        if spec.get("pg_id") is not None:
            bundle_key = self._bundle_key(
                (bytes(spec["pg_id"]), spec.get("bundle_index"))
            )
            if bundle_key is None:
                raise RuntimeError("placement group bundle not on this node")
            bundle = self.bundles[bundle_key]
            if not self._fits(bundle["avail"], res):
                raise RuntimeError(
                    f"bundle {bundle_key} lacks resources for actor {res}"
                )
            for k, v in res.items():
                bundle["avail"][k] = bundle["avail"].get(k, 0) - v
        elif not self._fits(self.avail, res):
            raise RuntimeError(f"node {self.node_name}: insufficient resources {res}")
        ngpu = float(res.get("GPU", 0))
        gpu_alloc = self._gpu_alloc(ngpu)
        if gpu_alloc is None:
            if bundle_key is not None:
                for k, v in res.items():
                    bundle["avail"][k] = bundle["avail"].get(k, 0) + v
            raise RuntimeError(
                f"node {self.node_name}: no GPU device has {ngpu} free "
                "(fragmented fractional occupancy)"
            )
        gpu_ids = [d for d, _ in gpu_alloc]
        if bundle_key is None:
            for k, v in res.items():
                self.avail[k] = self.avail.get(k, 0) - v
        if gpu_ids:
            spec.setdefault("env_vars", {})
            spec["env_vars"] = dict(spec.get("env_vars") or {})
            ids_str = ",".join(map(str, gpu_ids))
            spec["env_vars"]["RAY_AMD_GPU_IDS"] = ids_str
        w = self._spawn_worker(actor_spec=spec)
        w.gpu_ids = gpu_ids
        w.gpu_alloc = gpu_alloc
        w.actor_resources = res
        w.actor_bundle = bundle_key
        fut = asyncio.get_running_loop().create_future()
        self._actor_start_futs[actor_id] = fut
        try:
            r = await asyncio.wait_for(fut, 120.0)
        except asyncio.TimeoutError:
            try:
                w.proc.kill()
            except Exception:
                pass
            raise RuntimeError("actor start timed out")
        finally:
            self._actor_start_futs.pop(actor_id, None)
        if r.get("error"):
            raise RuntimeError(r["error"])
        return {"addr": r["addr"], "pid": w.pid}

    def h_actor_ready(self, conn, p):
        fut = self._actor_start_futs.get(bytes(p["actor_id"]))
        if fut is not None and not fut.done():
            fut.set_result({"addr": p["addr"]})
        return True

    def h_actor_failed(self, conn, p):
        fut = self._actor_start_futs.get(bytes(p["actor_id"]))
        if fut is not None and not fut.done():
            fut.set_result({"error": p.get("error", "actor init failed")})
        return True

    async def h_kill_worker(self, conn, p):
        addr = p["addr"]
        for w in self.workers.values():
            if w.addr == addr:
                try:
                    w.proc.kill()
                except Exception:
                    pass
                return True
        return False

    # ---------------- placement-group bundles (2PC participant) ----------------

    def h_reserve_bundle(self, conn, p):
        res = p["resources"]
        if not self._fits(self.avail, res):
            return {"ok": False}
        for k, v in res.items():
            self.avail[k] = self.avail.get(k, 0) - v
        self.bundles[(bytes(p["pg_id"]), p["bundle_index"])] = {
            "resources": dict(res),
            "avail": dict(res),
            "committed": False,
        }
        return {"ok": True}

    def h_commit_bundle(self, conn, p):
        b = self.bundles.get((bytes(p["pg_id"]), p["bundle_index"]))
        if b is not None:
            b["committed"] = True
        return {"ok": b is not None}

    def h_rollback_bundle(self, conn, p):
        b = self.bundles.pop((bytes(p["pg_id"]), p["bundle_index"]), None)
        if b is not None:
            for k, v in b["resources"].items():
                self.avail[k] = self.avail.get(k, 0) + v
        return True

    def h_remove_bundle(self, conn, p):
        return self.h_rollback_bundle(conn, p)

    # ---------------- object store ----------------

    def h_set_resource(self, conn, p):
        """Dynamic custom resources (reference:
        experimental/dynamic_resources.py ray.experimental.set_resource):
        adjust this node's capacity for one resource at runtime."""
        name = p["resource"]
        cap = float(p["capacity"])
        used = self.resources_total.get(name, 0.0) - self.avail.get(name, 0.0)
        if cap <= 0:
            self.resources_total.pop(name, None)
            self.avail.pop(name, None)
        else:
            self.resources_total[name] = cap
            self.avail[name] = max(0.0, cap - used)
        try:
            self.gcs.notify(
                "report_resources",
                {"node_id": self.node_id, "available": dict(self.avail),
                 "pending": self._pending_total(),
                 "total": dict(self.resources_total)},
            )
        except Exception:
            pass
        return {"ok": True, "total": self.resources_total.get(name, 0.0)}

    def h_seal_object(self, conn, p):
        self.store.seal(bytes(p["id"]), p["size"])
        return {"ok": True}

    async def h_wait_object(self, conn, p):
        oid = bytes(p["id"])
        if p.get("known_sealed") and not self.store.contains(oid):
            # caller knows the object was sealed here once: absence from
            # the table means it was freed/lost — fail fast so the owner
            # can reconstruct from lineage instead of stalling
            return {"ok": False}
        ok = await self.store.wait_sealed(oid, p.get("timeout", 60.0))
        if ok:
            self.store.ensure_local(oid)
        return {"ok": ok, "size": self.store.table.get(oid, [0])[0]}

    def h_debug_state(self, conn, p):
        """Per-handler event-loop stats + queue/pool gauges (reference:
        event_stats.cc periodic DebugString)."""
        return {
            "handler_stats": self.server.stats_table(),
            "pending_leases": self._pending_total(),
            "workers": len(self.workers),
            "idle_workers": len(self._idle_task_workers),
            "leases": len(self.leases),
            "pins": len(self._pins),
            "store": dict(zip(("objects", "used", "capacity"),
                              self.store.stats())),
        }

    def h_list_workers(self, conn, p):
        """Registered workers on this node (drives `ray_amd stack`)."""
        return [
            {"pid": pid, "addr": w.addr, "kind": w.kind,
             "idle": w.idle}
            for pid, w in self.workers.items()
            if w.addr and w.proc.poll() is None
        ]

    def h_free_objects(self, conn, p):
        ids_ = [bytes(i) for i in p["ids"]]
        now, deferred = [], []
        for oid in ids_:
            (deferred if self._pins.get(oid) else now).append(oid)
        if now:
            self.store.free(now)
        for oid in deferred:
            # pin-while-mapped (closes the round-1 recycle race,
            # reference: object pinning in plasma's lifecycle manager):
            # a reader still maps this segment; unlink when it unpins
            self._deferred_free.add(oid)

    # ---- mapping pins (readers register while they hold an mmap) ----

    def h_pin_object(self, conn, p):
        oid = bytes(p["id"])
        self._pins.setdefault(oid, set()).add(id(conn))
        self._conn_pins.setdefault(id(conn), set()).add(oid)

    def h_unpin_object(self, conn, p):
        self._unpin(bytes(p["id"]), id(conn))

    def _unpin(self, oid: bytes, conn_id: int):
        s = self._pins.get(oid)
        if s is not None:
            s.discard(conn_id)
            if not s:
                self._pins.pop(oid, None)
                if oid in self._deferred_free:
                    self._deferred_free.discard(oid)
                    self.store.free([oid])
        cp = self._conn_pins.get(conn_id)
        if cp is not None:
            cp.discard(oid)

    def h_try_recycle(self, conn, p):
        """Owner asks to take the freed segment into its hot pool.
        Granted only when no reader pins the mapping; otherwise the
        unlink is deferred to the last unpin and the pool gets
        nothing (correctness over recycling)."""
        oid = bytes(p["id"])
        if self._pins.get(oid):
            self._deferred_free.add(oid)
            return {"ok": False}
        ent = self.store.table.pop(oid, None)
        if ent is None or ent[1]:
            if ent is not None:
                self.store.table[oid] = ent
                self.store.free([oid])  # spilled: normal free
            return {"ok": False}
        self.store.used -= ent[0]
        return {"ok": True}

    # ---------------- object transfer ----------------
    #
    # Pull manager (reference: object_manager/pull_manager.h:52 —
    # prioritized bundles get > wait > task-args, dedup) + push-side
    # streaming (push_manager.h:28 — the source streams chunks with a
    # flow-control window instead of the round-trip-per-chunk fetch,
    # so cross-node bandwidth is not RTT-bound).

    PULL_PRIO_GET = 0
    PULL_PRIO_WAIT = 1
    PULL_PRIO_TASK_ARGS = 2
    MAX_CONCURRENT_PULLS = 8
    PUSH_WINDOW = 8  # chunks in flight before the source waits for acks

    async def h_pull_object(self, conn, p):
        """Pull an object from a remote node's store into ours.
        Deduplicated (concurrent pulls of one oid share a transfer) and
        prioritized; the transfer itself is source-paced push
        streaming."""
        oid = bytes(p["id"])
        if self.store.contains(oid):
            self.store.ensure_local(oid)
            return {"ok": True}
        st = self._pulls.get(oid)
        if st is None:
            st = self._pulls[oid] = {
                "fut": asyncio.get_running_loop().create_future(),
                "prio": int(p.get("prio", self.PULL_PRIO_TASK_ARGS)),
            }
            import heapq

            self._pull_seq += 1
            heapq.heappush(self._pull_heap,
                           (st["prio"], self._pull_seq, oid, p["src"]))
            asyncio.ensure_future(self._pull_pump())
        else:
            # a higher-priority duplicate bumps nothing in-flight but
            # is recorded for observability
            st["prio"] = min(st["prio"], int(p.get("prio", 2)))
        try:
            ok = await asyncio.wait_for(
                asyncio.shield(st["fut"]), p.get("timeout", 120.0)
            )
            return {"ok": ok}
        except asyncio.TimeoutError:
            return {"ok": False}

    async def _pull_pump(self):
        import heapq

        while self._pull_heap and self._pulls_active < self.MAX_CONCURRENT_PULLS:
            _, _, oid, src = heapq.heappop(self._pull_heap)
            st = self._pulls.get(oid)
            if st is None or st["fut"].done():
                continue
            self._pulls_active += 1
            asyncio.ensure_future(self._do_pull(oid, src, st))

    async def _do_pull(self, oid: bytes, src: str, st: dict):
        ok = False
        try:
            ok = await self._pull_streamed(oid, src)
        except Exception:
            traceback.print_exc()
        finally:
            self._pulls_active -= 1
            self._pulls.pop(oid, None)
            if not st["fut"].done():
                st["fut"].set_result(ok)
            asyncio.ensure_future(self._pull_pump())

    async def _pull_streamed(self, oid: bytes, src: str) -> bool:
        c = RpcClient()
        recv = {"writer": None, "size": None, "received": 0,
                "done": asyncio.get_running_loop().create_future()}

        def on_notify(method, payload):
            if method != "push_chunk" or bytes(payload["id"]) != oid:
                return
            try:
                if recv["writer"] is None:
                    recv["size"] = payload["size"]
                    recv["writer"] = store.ObjectWriter(
                        self.store.shm_dir, oid, max(payload["size"], 1)
                    )
                d = payload["data"]
                off = payload["off"]
                recv["writer"].view[off:off + len(d)] = d
                recv["received"] += len(d)
                c.notify("push_ack", {"id": oid, "received": recv["received"]})
                if recv["received"] >= recv["size"]:
                    if not recv["done"].done():
                        recv["done"].set_result(True)
            except Exception as e:
                if not recv["done"].done():
                    recv["done"].set_exception(e)

        try:
            await c.connect(src, retries=5)
            c.on_notify = on_notify
            r = await c.call("request_push", {"id": oid})
            if not r.get("ok"):
                return False
            if r.get("size") == 0:
                recv["writer"] = store.ObjectWriter(self.store.shm_dir, oid, 1)
                recv["size"] = 0
            else:
                await asyncio.wait_for(recv["done"], 120.0)
            recv["writer"].seal()
            self.store.seal(oid, recv["size"])
            return True
        except Exception:
            traceback.print_exc()
            return False
        finally:
            c.close()

    async def h_request_push(self, conn, p):
        """Source side: stream the object to the requester over this
        connection in CHUNK pieces, at most PUSH_WINDOW chunks ahead of
        the receiver's acks; deduplicate per (conn, oid)."""
        oid = bytes(p["id"])
        ok = await self.store.wait_sealed(oid, 30.0)
        if not ok:
            return {"ok": False}
        self.store.ensure_local(oid)
        key = (id(conn), oid)
        if key in self._pushes:
            return {"ok": True, "dup": True}
        size = self.store.table[oid][0]
        if size == 0:
            return {"ok": True, "size": 0}
        st = self._pushes[key] = {"acked": 0,
                                  "event": asyncio.Event()}

        async def _stream():
            try:
                path = store.shm_path(self.store.shm_dir, oid)
                with open(path, "rb") as f:
                    off = 0
                    while off < size:
                        while (off - st["acked"]) >= self.PUSH_WINDOW * CHUNK:
                            st["event"].clear()
                            try:
                                await asyncio.wait_for(st["event"].wait(),
                                                       60.0)
                            except asyncio.TimeoutError:
                                return
                        data = f.read(CHUNK)
                        from .protocol import MSG_NOTIFY

                        conn.send([MSG_NOTIFY, 0, "push_chunk",
                                   {"id": oid, "off": off, "size": size,
                                    "data": data}])
                        off += len(data)
            finally:
                self._pushes.pop(key, None)

        asyncio.ensure_future(_stream())
        return {"ok": True, "size": size}

    def h_push_ack(self, conn, p):
        st = self._pushes.get((id(conn), bytes(p["id"])))
        if st is not None:
            st["acked"] = max(st["acked"], int(p.get("received", 0)))
            st["event"].set()

    async def h_fetch_chunk(self, conn, p):
        oid = bytes(p["id"])
        ok = await self.store.wait_sealed(oid, 30.0)
        if not ok:
            return {"ok": False}
        self.store.ensure_local(oid)
        path = store.shm_path(self.store.shm_dir, oid)
        size = self.store.table[oid][0]
        with open(path, "rb") as f:
            f.seek(p["off"])
            data = f.read(p["len"])
        return {"ok": True, "size": size, "data": data}

    def h_object_stats(self, conn, p):
        n, used, cap = self.store.stats()
        return {"num_objects": n, "used": used, "capacity": cap}

    def h_node_info(self, conn, p):
        return {
            "node_id": self.node_id,
            "addr": self.addr,
            "resources_total": self.resources_total,
            "resources_available": self.avail,
            "num_workers": len(self.workers),
        }

    def h_ping(self, conn, p):
        return "pong"

    def h_report_task_events(self, conn, p):
        spans = p.get("spans")
        if spans:
            try:
                self.gcs.notify("timeline_events",
                                {"events": [{"span": sp} for sp in spans]})
            except Exception:
                pass
        self._task_events.extend(p.get("events", []))
        if len(self._task_events) > 100000:
            del self._task_events[:50000]
        evs = self._task_events
        if p.get("fetch"):
            return evs
        return len(evs)

    def shutdown_workers(self):
        for w in self.workers.values():
            try:
                w.proc.kill()
            except Exception:
                pass


def main():
    import argparse
    import json

    ap = argparse.ArgumentParser()
    ap.add_argument("--session-dir", required=True)
    ap.add_argument("--gcs", required=True)
    ap.add_argument("--resources", required=True, help="json dict")
    ap.add_argument("--node-name", default="")
    ap.add_argument("--labels", default="{}")
    ap.add_argument("--object-store-memory", type=int, default=0)
    ap.add_argument("--ready-file", default="")
    args = ap.parse_args()

    raylet = Raylet(
        args.session_dir,
        args.gcs,
        json.loads(args.resources),
        node_name=args.node_name,
        labels=json.loads(args.labels),
        object_store_memory=args.object_store_memory or None,
    )

    async def run():
        await raylet.start()
        if args.ready_file:
            # atomic write: the starter polls for existence
            tmp = args.ready_file + ".tmp"
            with open(tmp, "w") as f:
                f.write(raylet.addr + "\n" + raylet.node_id.hex())
            os.replace(tmp, args.ready_file)
        try:
            # on GCS loss: try to reconnect + re-register for 30s (GCS
            # restart with persisted state); exit if it stays gone
            while True:
                if raylet.gcs.connected:
                    await asyncio.sleep(0.5)
                    continue
                recovered = False
                deadline = time.time() + 30.0
                while time.time() < deadline:
                    try:
                        c = RpcClient()
                        await c.connect(raylet.gcs_addr, retries=5)
                        raylet.gcs = c
                        await raylet.gcs.call(
                            "register_node",
                            {
                                "node_id": raylet.node_id,
                                "addr": raylet.addr,
                                "resources": raylet.resources_total,
                                "labels": raylet.labels,
                            },
                        )
                        recovered = True
                        break
                    except Exception:
                        await asyncio.sleep(0.5)
                if not recovered:
                    return
        finally:
            raylet.shutdown_workers()

    try:
        asyncio.run(run())
    finally:
        raylet.shutdown_workers()


if __name__ == "__main__":
    main()
