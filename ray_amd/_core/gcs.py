"""GCS — the cluster control plane process.

Feature counterpart of the reference's gcs_server (src/ray/gcs/gcs_server.h:99):
node membership, actor lifecycle FSM + placement (gcs/actor/gcs_actor_manager.h:94),
placement groups with 2-phase reserve/commit (gcs_placement_group_manager.h:51),
namespaced KV (gcs_kv_manager.cc), job ids, and actor-state watch
(pubsub). Re-designed as one asyncio process over the msgpack RPC layer.
"""
from __future__ import annotations

import asyncio
import time
import traceback
from typing import Dict, List, Optional

from . import ids
from .protocol import RpcClient, RpcServer

ACTOR_PENDING = "PENDING_CREATION"
ACTOR_ALIVE = "ALIVE"
ACTOR_RESTARTING = "RESTARTING"
ACTOR_DEAD = "DEAD"


class NodeInfo:
    def __init__(self, node_id: bytes, addr: str, resources: Dict[str, float], labels):
        self.node_id = node_id
        self.addr = addr
        self.resources_total = dict(resources)
        self.resources_available = dict(resources)
        self.labels = labels or {}
        self.alive = True
        self.draining = False  # graceful drain: no new placements
        self.client: Optional[RpcClient] = None
        self.start_time = time.time()

    def adjust(self, delta: Dict[str, float], sign: float = 1.0):
        """Optimistic cache adjustment, clamped to [0, total] — raylet
        reports are the truth and overwrite this within a second."""
        for k, v in delta.items():
            cur = self.resources_available.get(k, 0.0) + sign * v
            hi = self.resources_total.get(k, cur)
            self.resources_available[k] = min(max(cur, 0.0), hi)


class ActorInfo:
    def __init__(self, actor_id: bytes, spec: dict):
        self.actor_id = actor_id
        self.spec = spec
        self.name = spec.get("name")
        self.namespace = spec.get("namespace", "default")
        self.state = ACTOR_PENDING
        self.addr: Optional[str] = None
        self.node_id: Optional[bytes] = None
        self.num_restarts = 0
        self.death_cause: Optional[str] = None
        self.waiters: List[asyncio.Future] = []


class PlacementGroupInfo:
    def __init__(self, pg_id: bytes, bundles: List[dict], strategy: str, name: str):
        self.pg_id = pg_id
        self.bundles = bundles
        self.strategy = strategy
        self.name = name
        self.state = "PENDING"
        self.bundle_nodes: List[Optional[bytes]] = [None] * len(bundles)
        self.waiters: List[asyncio.Future] = []


class GcsServer:
    """Set persist_path to journal KV + actor + PG state; a restarted
    GCS on the same socket replays it (reference: gcs/store_client Redis
    persistence + gcs_init_data replay)."""

    def __init__(self, sock_path: str, persist_path: str = ""):
        self.sock_path = sock_path
        self.persist_path = persist_path
        self._storage = None
        if persist_path:
            from .gcs_storage import open_storage

            self._storage = open_storage(persist_path)
        self._dirty = False
        self.server = RpcServer()
        self.kv: Dict[str, Dict[bytes, bytes]] = {}
        self.nodes: Dict[bytes, NodeInfo] = {}
        self.actors: Dict[bytes, ActorInfo] = {}
        self.named_actors: Dict[tuple, bytes] = {}
        self.pgs: Dict[bytes, PlacementGroupInfo] = {}
        self.job_counter = 0
        self._proto_node: Dict[int, bytes] = {}  # id(proto) -> node_id
        # pub/sub topic bus (reference: gcs_publisher/gcs_subscriber —
        # log/error/node/actor channels): channel -> subscribed conns
        self._subs: Dict[str, set] = {}
        for m in (
            "kv_put kv_get kv_del kv_keys kv_exists register_node node_table "
            "report_resources register_actor resolve_actor actor_exit "
            "kill_actor list_actors next_job_id create_pg pg_wait_ready "
            "remove_pg pg_table ping timeline_events drain_node "
            "subscribe unsubscribe publish debug_state"
        ).split():
            self.server.route(m, getattr(self, "h_" + m))
        self.server.on_conn_lost = self._conn_lost
        self._timeline: List[dict] = []

    async def start(self):
        if self.persist_path:
            self._restore()
            asyncio.ensure_future(self._persist_loop())
        import os as _os

        try:
            _os.unlink(self.sock_path)
        except OSError:
            pass
        node_ip = _os.environ.get("RAY_AMD_NODE_IP")
        if node_ip:
            # multi-machine mode: listen on TCP, advertise the address
            # through the sock_path as a regular file (node.start_gcs
            # reads it back)
            port = await self.server.start_tcp(node_ip, 0)
            self.addr = f"tcp:{node_ip}:{port}"
            with open(self.sock_path + ".tmp", "w") as f:
                f.write(self.addr)
            _os.rename(self.sock_path + ".tmp", self.sock_path)
        else:
            await self.server.start_unix(self.sock_path)
            self.addr = "unix:" + self.sock_path

    # ---------- persistence ----------
    def _snapshot(self) -> bytes:
        import pickle

        return pickle.dumps(
            {
                "kv": self.kv,
                "named_actors": self.named_actors,
                "actor_specs": {
                    aid: a.spec for aid, a in self.actors.items()
                    if a.state != ACTOR_DEAD
                },
                "job_counter": self.job_counter,
            }
        )

    def _restore(self):
        import pickle

        from .gcs_storage import decode_op

        blob, ops = self._storage.load()
        snap = {}
        if blob is not None:
            try:
                snap = pickle.loads(blob)
            except Exception:
                snap = {}
        self.kv = snap.get("kv", {})
        self.named_actors = snap.get("named_actors", {})
        self.job_counter = snap.get("job_counter", 0)
        # replay the synchronous KV journal on top of the snapshot
        # (sqlite backend: acknowledged writes survive kill -9)
        for raw in ops:
            try:
                op = decode_op(raw)
            except Exception:
                continue
            if op[0] == "kv_put":
                _, ns, key, value = op
                self.kv.setdefault(ns, {})[key] = value
            elif op[0] == "kv_del":
                _, ns, key = op
                self.kv.get(ns, {}).pop(key, None)
            elif op[0] == "actor_put":
                snap.setdefault("actor_specs", {})[op[1]] = op[2]
            elif op[0] == "actor_del":
                snap.setdefault("actor_specs", {}).pop(op[1], None)
        # actors come back PENDING and are rescheduled once a raylet
        # re-registers (restart-based recovery)
        for aid, spec in snap.get("actor_specs", {}).items():
            a = ActorInfo(aid, spec)
            self.actors[aid] = a
            if a.name:  # journaled actors re-register their name
                self.named_actors.setdefault((a.namespace, a.name), aid)
            asyncio.ensure_future(self._schedule_actor(a))

    async def _persist_loop(self):
        while True:
            await asyncio.sleep(0.5)
            if not self._dirty:
                continue
            self._dirty = False
            try:
                self._storage.save_snapshot(self._snapshot())
            except Exception:
                pass

    def _journal(self, *op):
        if self._storage is None:
            return
        from .gcs_storage import encode_op

        try:
            self._storage.journal(encode_op(*op))
        except Exception:
            pass

    # ---------- KV ----------
    def h_kv_put(self, conn, p):
        ns = self.kv.setdefault(p.get("ns", ""), {})
        key = p["key"]
        exists = key in ns
        if p.get("overwrite", True) or not exists:
            ns[key] = p["value"]
            self._dirty = True
            if self._storage is not None:
                from .gcs_storage import encode_op

                self._storage.journal(
                    encode_op("kv_put", p.get("ns", ""), key, p["value"]))
            return not exists
        return False

    def h_kv_get(self, conn, p):
        return self.kv.get(p.get("ns", ""), {}).get(p["key"])

    def h_kv_del(self, conn, p):
        hit = self.kv.get(p.get("ns", ""), {}).pop(p["key"], None) is not None
        if hit and self._storage is not None:
            from .gcs_storage import encode_op

            self._storage.journal(
                encode_op("kv_del", p.get("ns", ""), p["key"]))
        return hit

    def h_kv_exists(self, conn, p):
        return p["key"] in self.kv.get(p.get("ns", ""), {})

    def h_kv_keys(self, conn, p):
        prefix = p.get("prefix", b"")
        return [k for k in self.kv.get(p.get("ns", ""), {}) if k.startswith(prefix)]

    # ---------- nodes ----------
    async def h_register_node(self, conn, p):
        node_id = p["node_id"]
        info = NodeInfo(node_id, p["addr"], p["resources"], p.get("labels"))
        client = RpcClient()
        await client.connect(p["addr"])
        info.client = client
        self.nodes[node_id] = info
        self._proto_node[id(conn)] = node_id
        return {"ok": True}

    def h_node_table(self, conn, p):
        return [
            {
                "node_id": n.node_id,
                "addr": n.addr,
                "alive": n.alive,
                "draining": n.draining,
                "resources_total": n.resources_total,
                "resources_available": n.resources_available,
                "labels": n.labels,
                "pending": getattr(n, "pending", 0),
                "pending_shapes": getattr(n, "pending_shapes", []),
            }
            for n in self.nodes.values()
        ]

    def h_report_resources(self, conn, p):
        n = self.nodes.get(p["node_id"])
        if n is not None:
            n.resources_available = p["available"]
            n.pending = p.get("pending", 0)
            n.pending_shapes = p.get("pending_shapes") or []
            if p.get("total") is not None:  # dynamic resource change
                n.resources_total = p["total"]

    def h_debug_state(self, conn, p):
        return {
            "handler_stats": self.server.stats_table(),
            "nodes": len(self.nodes),
            "actors": len(self.actors),
            "pgs": len(self.pgs),
        }

    def h_ping(self, conn, p):
        return "pong"

    def h_timeline_events(self, conn, p):
        evs = p.get("events")
        if evs:
            self._timeline.extend(evs)
            return len(self._timeline)
        return self._timeline

    def _conn_lost(self, proto, exc):
        for subs in self._subs.values():
            subs.discard(proto)
        node_id = self._proto_node.pop(id(proto), None)
        if node_id is not None and node_id in self.nodes:
            asyncio.ensure_future(self._on_node_death(node_id))

    # ---------- pub/sub ----------

    def h_subscribe(self, conn, p):
        self._subs.setdefault(p["channel"], set()).add(conn)
        return True

    def h_unsubscribe(self, conn, p):
        subs = self._subs.get(p["channel"])
        if subs is not None:
            subs.discard(conn)
        return True

    def h_publish(self, conn, p):
        from .protocol import MSG_NOTIFY

        ch = p["channel"]
        n = 0
        for proto in list(self._subs.get(ch, ())):
            if proto.transport is None or proto.transport.is_closing():
                self._subs[ch].discard(proto)
                continue
            proto.send([MSG_NOTIFY, 0, "pubsub",
                        {"channel": ch, "data": p["data"]}])
            n += 1
        return n

    async def h_drain_node(self, conn, p):
        """Multi-phase node drain (reference: DrainNode RPC + the
        raylet shutdown FSM): graceful mode marks the node DRAINING —
        the scheduler stops placing work there immediately — then
        waits for its leases to finish (resources fully free) or the
        deadline before declaring it dead. Non-graceful keeps the old
        immediate-death behavior."""
        node_id = bytes(p["node_id"])
        if not p.get("graceful"):
            await self._on_node_death(node_id)
            return True
        n = self.nodes.get(node_id)
        if n is None or not n.alive:
            return False
        n.draining = True
        deadline = time.time() + float(p.get("deadline_s", 30.0))

        async def _watch():
            while time.time() < deadline:
                cur = self.nodes.get(node_id)
                if cur is None or not cur.alive:
                    return
                busy = any(
                    cur.resources_available.get(k, 0) + 1e-9 < v
                    for k, v in cur.resources_total.items())
                if not busy and getattr(cur, "pending", 0) == 0:
                    break
                await asyncio.sleep(0.1)
            await self._on_node_death(node_id)

        asyncio.ensure_future(_watch())
        return True

    async def _on_node_death(self, node_id: bytes):
        n = self.nodes.get(node_id)
        if n is None or not n.alive:
            return
        n.alive = False
        for a in list(self.actors.values()):
            if a.node_id == node_id and a.state == ACTOR_ALIVE:
                await self._on_actor_exit(a, "node died", expected=False)

    # ---------- scheduling helpers ----------
    def _alive_nodes(self) -> List[NodeInfo]:
        # draining nodes take no NEW placements (existing work drains)
        return [n for n in self.nodes.values()
                if n.alive and not n.draining]

    def _fits(self, node: NodeInfo, req: Dict[str, float]) -> bool:
        for k, v in req.items():
            if v > 0 and node.resources_available.get(k, 0.0) + 1e-9 < v:
                return False
        return True

    def _pick_node(self, req: Dict[str, float], strategy: str = "hybrid",
                   exclude=(), soft_affinity: Optional[bytes] = None,
                   node_affinity=None, label_selector=None) -> Optional[NodeInfo]:
        """Hybrid policy (reference: policy/hybrid_scheduling_policy.h:28):
        sort feasible nodes by load and pick uniformly among the top k
        (top-k randomization avoids herding every scheduler decision
        onto one node; reference scheduler_top_k_fraction). Node
        affinity / label selectors constrain the candidate set first
        (reference: node_affinity_scheduling_policy.cc, node-label
        policy)."""
        cands = [n for n in self._alive_nodes() if n.node_id not in exclude and self._fits(n, req)]
        if node_affinity is not None:
            target_hex, soft = node_affinity
            pinned = [n for n in cands if n.node_id.hex() == target_hex]
            if pinned:
                return pinned[0]
            if not soft:
                return None  # hard affinity: wait for that node or fail
        if label_selector:
            hard = dict(label_selector.get("hard") or {})
            soft_l = dict(label_selector.get("soft") or {})
            cands = [
                n for n in cands
                if all((n.labels or {}).get(k) == v for k, v in hard.items())
            ]
            soft_match = [
                n for n in cands
                if all((n.labels or {}).get(k) == v for k, v in soft_l.items())
            ]
            if soft_match:
                cands = soft_match
        if not cands:
            return None
        if soft_affinity is not None:
            for n in cands:
                if n.node_id == soft_affinity:
                    return n

        def load(n: NodeInfo):
            t = n.resources_total.get("CPU", 1.0) or 1.0
            return 1.0 - n.resources_available.get("CPU", 0.0) / t

        cands.sort(key=load)
        from .._config import config as _cfgmod

        k = max(
            _cfgmod.scheduler_top_k_absolute,
            int(len(cands) * _cfgmod.scheduler_top_k_fraction),
        )
        k = min(k, len(cands))
        import random as _random

        return _random.choice(cands[:k])

    # ---------- actors ----------
    async def h_register_actor(self, conn, p):
        actor_id = p["actor_id"]
        a = ActorInfo(actor_id, p)
        if a.name:
            key = (a.namespace, a.name)
            if key in self.named_actors:
                other = self.actors.get(self.named_actors[key])
                if other is not None and other.state != ACTOR_DEAD:
                    if p.get("get_if_exists"):
                        return {"existing": self.named_actors[key]}
                    raise ValueError(f"actor name {a.name!r} already taken")
            self.named_actors[key] = actor_id
        self.actors[actor_id] = a
        self._dirty = True
        self._journal("actor_put", actor_id, a.spec)
        asyncio.ensure_future(self._schedule_actor(a))
        return {"existing": None}

    async def _schedule_actor(self, a: ActorInfo):
        req = dict(a.spec.get("resources", {}))
        in_pg = a.spec.get("pg_id") is not None
        deadline = time.time() + 300.0
        while a.state in (ACTOR_PENDING, ACTOR_RESTARTING):
            if in_pg:
                # bundle resources were reserved at PG creation; the
                # actor MUST land on the bundle's node and must not be
                # double-counted against node availability
                pg_node = a.spec.get("pg_node")
                node = None
                if pg_node is not None:
                    n = self.nodes.get(bytes(pg_node))
                    node = n if n is not None and n.alive else None
                else:
                    pg = self.pgs.get(bytes(a.spec["pg_id"]))
                    if pg is not None:
                        for nid in pg.bundle_nodes:
                            if nid and self.nodes.get(nid) and self.nodes[nid].alive:
                                node = self.nodes[nid]
                                break
            else:
                node = self._pick_node(
                    req,
                    strategy=a.spec.get("scheduling_strategy", "hybrid"),
                    soft_affinity=a.spec.get("pg_node"),
                    node_affinity=a.spec.get("node_affinity"),
                    label_selector=a.spec.get("label_selector"),
                )
            if node is None:
                if time.time() > deadline:
                    await self._fail_actor(a, "resources unavailable for actor")
                    return
                await asyncio.sleep(0.05)
                continue
            try:
                r = await node.client.call(
                    "start_actor", {"actor_id": a.actor_id, "spec": a.spec}
                )
                a.node_id = node.node_id
                a.addr = r["addr"]
                a.state = ACTOR_ALIVE
                if not in_pg:
                    node.adjust(req, -1.0)
                for f in a.waiters:
                    if not f.done():
                        f.set_result(None)
                a.waiters.clear()
                return
            except Exception:
                traceback.print_exc()
                if time.time() > deadline:
                    await self._fail_actor(a, "actor start failed:\n" + traceback.format_exc())
                    return
                await asyncio.sleep(0.2)

    async def _fail_actor(self, a: ActorInfo, cause: str):
        a.state = ACTOR_DEAD
        a.death_cause = cause
        self._journal("actor_del", a.actor_id)
        for f in a.waiters:
            if not f.done():
                f.set_result(None)
        a.waiters.clear()

    async def h_resolve_actor(self, conn, p):
        aid = p.get("actor_id")
        if aid is None:
            key = (p.get("namespace", "default"), p["name"])
            aid = self.named_actors.get(key)
            if aid is None:
                return {"state": "NOT_FOUND"}
        a = self.actors.get(aid)
        if a is None:
            return {"state": "NOT_FOUND"}
        if p.get("wait") and a.state in (ACTOR_PENDING, ACTOR_RESTARTING):
            fut = asyncio.get_running_loop().create_future()
            a.waiters.append(fut)
            try:
                await asyncio.wait_for(fut, p.get("timeout", 120.0))
            except asyncio.TimeoutError:
                pass
        return {
            "state": a.state,
            "actor_id": aid,
            "addr": a.addr,
            "node_id": a.node_id,
            "death_cause": a.death_cause,
            "spec_kv_key": a.spec.get("spec_kv_key"),
            "working_dir": a.spec.get("working_dir"),
            "py_modules": a.spec.get("py_modules"),
            "tensor_transport": a.spec.get("tensor_transport"),
        }

    # (profiler wrapper rides on the raylet start_actor spec, not the
    # worker resolve path)

    async def h_actor_exit(self, conn, p):
        a = self.actors.get(p["actor_id"])
        if a is None:
            return
        await self._on_actor_exit(a, p.get("cause", "actor exited"), p.get("expected", True))

    async def _on_actor_exit(self, a: ActorInfo, cause: str, expected: bool):
        node = self.nodes.get(a.node_id) if a.node_id else None
        if node is not None and a.state == ACTOR_ALIVE and not a.spec.get("pg_id"):
            node.adjust(a.spec.get("resources", {}), +1.0)
        if (not expected) and a.num_restarts < a.spec.get("max_restarts", 0):
            a.num_restarts += 1
            a.state = ACTOR_RESTARTING
            a.addr = None
            asyncio.ensure_future(self._schedule_actor(a))
        else:
            a.state = ACTOR_DEAD
            a.death_cause = cause
            self._journal("actor_del", a.actor_id)
            if a.name and self.named_actors.get((a.namespace, a.name)) == a.actor_id:
                del self.named_actors[(a.namespace, a.name)]
            for f in a.waiters:
                if not f.done():
                    f.set_result(None)
            a.waiters.clear()

    async def h_kill_actor(self, conn, p):
        a = self.actors.get(p["actor_id"])
        if a is None or a.state == ACTOR_DEAD:
            return False
        if a.addr and a.state == ACTOR_ALIVE:
            node = self.nodes.get(a.node_id)
            if node is not None and node.client is not None:
                try:
                    await node.client.call(
                        "kill_worker", {"addr": a.addr, "no_restart": p.get("no_restart", True)}
                    )
                except Exception:
                    pass
        if p.get("no_restart", True):
            a.spec["max_restarts"] = 0
        await self._on_actor_exit(a, "ray.kill", expected=p.get("no_restart", True))
        return True

    def h_list_actors(self, conn, p):
        out = []
        for a in self.actors.values():
            out.append(
                {
                    "actor_id": a.actor_id,
                    "class_name": a.spec.get("class_name"),
                    "name": a.name,
                    "namespace": a.namespace,
                    "state": a.state,
                    "node_id": a.node_id,
                    "pid": a.spec.get("pid"),
                    "num_restarts": a.num_restarts,
                }
            )
        return out

    def h_next_job_id(self, conn, p):
        self.job_counter += 1
        return self.job_counter

    # ---------- placement groups (2PC: reserve on each raylet, commit) ----------
    async def h_create_pg(self, conn, p):
        pg = PlacementGroupInfo(p["pg_id"], p["bundles"], p["strategy"], p.get("name", ""))
        self.pgs[pg.pg_id] = pg
        asyncio.ensure_future(self._schedule_pg(pg))
        return {"pg_id": pg.pg_id}

    async def _schedule_pg(self, pg: PlacementGroupInfo):
        deadline = time.time() + 300.0
        while pg.state == "PENDING":
            plan = self._plan_pg(pg)
            if plan is None:
                if time.time() > deadline:
                    pg.state = "INFEASIBLE"
                    break
                await asyncio.sleep(0.05)
                continue
            reserved = []
            ok = True
            for idx, node in enumerate(plan):
                try:
                    r = await node.client.call(
                        "reserve_bundle",
                        {"pg_id": pg.pg_id, "bundle_index": idx, "resources": pg.bundles[idx]},
                    )
                    if not r.get("ok"):
                        ok = False
                        break
                    reserved.append((idx, node))
                except Exception:
                    ok = False
                    break
            if ok:
                for idx, node in reserved:
                    await node.client.call(
                        "commit_bundle", {"pg_id": pg.pg_id, "bundle_index": idx}
                    )
                    pg.bundle_nodes[idx] = node.node_id
                    node.adjust(pg.bundles[idx], -1.0)
                pg.state = "CREATED"
                break
            for idx, node in reserved:
                try:
                    await node.client.call(
                        "rollback_bundle", {"pg_id": pg.pg_id, "bundle_index": idx}
                    )
                except Exception:
                    pass
            await asyncio.sleep(0.05)
            if time.time() > deadline:
                pg.state = "INFEASIBLE"
                break
        for f in pg.waiters:
            if not f.done():
                f.set_result(None)
        pg.waiters.clear()

    def _plan_pg(self, pg: PlacementGroupInfo) -> Optional[List[NodeInfo]]:
        nodes = self._alive_nodes()
        if not nodes:
            return None
        avail = {n.node_id: dict(n.resources_available) for n in nodes}

        def fits(nid, req):
            a = avail[nid]
            return all(a.get(k, 0) + 1e-9 >= v for k, v in req.items())

        def take(nid, req):
            for k, v in req.items():
                avail[nid][k] = avail[nid].get(k, 0) - v

        plan: List[NodeInfo] = []
        strat = pg.strategy
        if strat in ("STRICT_PACK", "PACK"):
            # try to pack everything on one node first
            for n in sorted(nodes, key=lambda n: -n.resources_available.get("CPU", 0)):
                a = dict(avail[n.node_id])
                if all(self._consume(a, b) for b in pg.bundles):
                    return [n] * len(pg.bundles)
            if strat == "STRICT_PACK":
                return None
        if strat in ("STRICT_SPREAD", "SPREAD"):
            used = set()
            for b in pg.bundles:
                cand = [
                    n for n in nodes
                    if fits(n.node_id, b) and (n.node_id not in used or strat == "SPREAD")
                ]
                cand.sort(key=lambda n: (n.node_id in used, -avail[n.node_id].get("CPU", 0)))
                if not cand:
                    return None
                n = cand[0]
                take(n.node_id, b)
                used.add(n.node_id)
                plan.append(n)
            return plan
        # PACK fallback / default: greedy best-fit
        plan = []
        for b in pg.bundles:
            cand = [n for n in nodes if fits(n.node_id, b)]
            if not cand:
                return None
            cand.sort(key=lambda n: -avail[n.node_id].get("CPU", 0))
            take(cand[0].node_id, b)
            plan.append(cand[0])
        return plan

    @staticmethod
    def _consume(avail: Dict[str, float], req: Dict[str, float]) -> bool:
        if all(avail.get(k, 0) + 1e-9 >= v for k, v in req.items()):
            for k, v in req.items():
                avail[k] = avail.get(k, 0) - v
            return True
        return False

    async def h_pg_wait_ready(self, conn, p):
        pg = self.pgs.get(p["pg_id"])
        if pg is None:
            raise ValueError("no such placement group")
        if pg.state == "PENDING":
            fut = asyncio.get_running_loop().create_future()
            pg.waiters.append(fut)
            try:
                await asyncio.wait_for(fut, p.get("timeout", 120.0))
            except asyncio.TimeoutError:
                pass
        return {"state": pg.state, "bundle_nodes": pg.bundle_nodes}

    async def h_remove_pg(self, conn, p):
        pg = self.pgs.pop(p["pg_id"], None)
        if pg is None:
            return False
        for idx, nid in enumerate(pg.bundle_nodes):
            node = self.nodes.get(nid) if nid else None
            if node is not None and node.alive:
                try:
                    await node.client.call(
                        "remove_bundle", {"pg_id": pg.pg_id, "bundle_index": idx}
                    )
                    node.adjust(pg.bundles[idx], +1.0)
                except Exception:
                    pass
        return True

    def h_pg_table(self, conn, p):
        return [
            {"pg_id": g.pg_id, "name": g.name, "state": g.state, "strategy": g.strategy,
             "bundles": g.bundles, "bundle_nodes": g.bundle_nodes}
            for g in self.pgs.values()
        ]


def main():
    import sys

    sock = sys.argv[1]
    persist = sys.argv[2] if len(sys.argv) > 2 else ""
    gcs = GcsServer(sock, persist)

    async def run():
        await gcs.start()
        had_nodes = False
        idle_since = None
        while True:
            await asyncio.sleep(1.0)
            alive = any(n.alive for n in gcs.nodes.values())
            if alive:
                had_nodes = True
                idle_since = None
            elif had_nodes:
                idle_since = idle_since or time.time()
                if time.time() - idle_since > 10.0:
                    return  # all raylets gone: session over

    asyncio.run(run())


if __name__ == "__main__":
    main()
