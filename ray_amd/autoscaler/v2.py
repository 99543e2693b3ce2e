"""Autoscaler v2: instance-manager FSM + demand bin-packing scheduler
(reference: autoscaler/v2/ — autoscaler.py, scheduler.py,
instance_manager/; GCS autoscaler state from
gcs_autoscaler_state_manager.cc).

Demand = pending lease SHAPES reported by every raylet (ray_syncer
reports carry them) + actors stuck PENDING at the GCS. The scheduler
first fits each demand shape onto live nodes' availability, then
bin-packs the remainder into new instances chosen from the node-type
table; the instance manager drives each instance through
QUEUED -> REQUESTED -> RUNNING -> TERMINATING and reconciles with the
GCS node table. Providers are pluggable; LocalNodeProvider launches
raylets on this machine (the FakeMultiNodeProvider role).
"""
from __future__ import annotations

import threading
import time
import uuid
from typing import Dict, List, Optional

QUEUED = "QUEUED"
REQUESTED = "REQUESTED"
RUNNING = "RUNNING"
TERMINATING = "TERMINATING"
TERMINATED = "TERMINATED"


class NodeType:
    def __init__(self, name: str, resources: Dict[str, float],
                 min_workers: int = 0, max_workers: int = 10):
        self.name = name
        self.resources = dict(resources)
        self.min_workers = min_workers
        self.max_workers = max_workers


class Instance:
    def __init__(self, node_type: str):
        self.id = uuid.uuid4().hex[:8]
        self.node_type = node_type
        self.status = QUEUED
        self.handle = None           # provider handle
        self.node_id_hex: Optional[str] = None
        self.launched_at = time.time()
        self.idle_since: Optional[float] = None


class NodeProvider:
    """Plugin ABC (reference: autoscaler NodeProvider)."""

    def create_node(self, node_type: NodeType, labels: dict):
        raise NotImplementedError

    def terminate_node(self, handle) -> None:
        raise NotImplementedError


class LocalNodeProvider(NodeProvider):
    """Starts extra raylets on this machine (FakeMultiNode role)."""

    def __init__(self, cluster):
        self.cluster = cluster

    def create_node(self, node_type: NodeType, labels: dict):
        return self.cluster.add_node(resources=dict(node_type.resources),
                                     labels=labels)

    def terminate_node(self, handle) -> None:
        self.cluster.remove_node(handle)


class AutoscalerV2:
    def __init__(self, provider: NodeProvider,
                 node_types: List[NodeType], *,
                 idle_timeout_s: float = 60.0, poll_s: float = 0.25,
                 upscale_after_s: float = 0.3):
        self.provider = provider
        self.node_types = {t.name: t for t in node_types}
        self.idle_timeout_s = idle_timeout_s
        self.poll_s = poll_s
        self.upscale_after_s = upscale_after_s
        self.instances: List[Instance] = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._demand_since: Optional[float] = None

    # ---------------- state readers ----------------

    def _rt(self):
        from ray_amd._core import runtime as rtmod

        return rtmod.global_runtime()

    def _cluster_state(self):
        rt = self._rt()
        nodes = rt.gcs_call("node_table", {})
        actors = rt.gcs_call("list_actors", {})
        pending_shapes: List[dict] = []
        for n in nodes:
            if not n["alive"]:
                continue
            for shape, count in n.get("pending_shapes") or []:
                pending_shapes.extend([dict(shape)] * int(count))
        for a in actors:
            if a.get("state") == "PENDING_CREATION" and a.get("resources"):
                pending_shapes.append(dict(a["resources"]))
        return nodes, pending_shapes

    # ---------------- scheduler ----------------

    def _plan(self, nodes, demand: List[dict]) -> Dict[str, int]:
        """Bin-pack unmet demand into new instances per node type.
        Returns {type_name: count_to_launch}."""
        avail = [dict(n["resources_available"]) for n in nodes if n["alive"]]

        def fit(pool, shape):
            for k, v in shape.items():
                if v > 0 and pool.get(k, 0) + 1e-9 < v:
                    return False
            return True

        def take(pool, shape):
            for k, v in shape.items():
                pool[k] = pool.get(k, 0) - v

        unmet = []
        for shape in demand:
            for pool in avail:
                if fit(pool, shape):
                    take(pool, shape)
                    break
            else:
                unmet.append(shape)

        launches: Dict[str, int] = {}
        live_counts: Dict[str, int] = {}
        for inst in self.instances:
            if inst.status in (QUEUED, REQUESTED, RUNNING):
                live_counts[inst.node_type] = (
                    live_counts.get(inst.node_type, 0) + 1)
        new_pools: List[tuple] = []  # (type_name, pool)
        for shape in unmet:
            placed = False
            for tname, pool in new_pools:
                if fit(pool, shape):
                    take(pool, shape)
                    placed = True
                    break
            if placed:
                continue
            for t in self.node_types.values():
                total = (live_counts.get(t.name, 0)
                         + launches.get(t.name, 0))
                if total >= t.max_workers:
                    continue
                if fit(dict(t.resources), shape):
                    pool = dict(t.resources)
                    take(pool, shape)
                    new_pools.append((t.name, pool))
                    launches[t.name] = launches.get(t.name, 0) + 1
                    break
        # honor min_workers
        for t in self.node_types.values():
            have = live_counts.get(t.name, 0) + launches.get(t.name, 0)
            if have < t.min_workers:
                launches[t.name] = (launches.get(t.name, 0)
                                    + (t.min_workers - have))
        return launches

    # ---------------- instance manager ----------------

    def _reconcile(self, nodes):
        by_label = {}
        for n in nodes:
            lab = (n.get("labels") or {}).get("ray_amd_instance")
            if lab:
                by_label[lab] = n
        for inst in self.instances:
            if inst.status == REQUESTED:
                n = by_label.get(inst.id)
                if n is not None and n["alive"]:
                    inst.status = RUNNING
                    inst.node_id_hex = bytes(n["node_id"]).hex()
            elif inst.status == RUNNING:
                n = by_label.get(inst.id)
                if n is None or not n["alive"]:
                    inst.status = TERMINATED
                    continue
                total = n["resources_total"]
                availr = n["resources_available"]
                busy = any(
                    availr.get(k, 0) + 1e-9 < v for k, v in total.items()
                ) or n.get("pending", 0) > 0
                if busy:
                    inst.idle_since = None
                elif inst.idle_since is None:
                    inst.idle_since = time.time()
                elif time.time() - inst.idle_since > self.idle_timeout_s:
                    t = self.node_types[inst.node_type]
                    live = sum(1 for i in self.instances
                               if i.node_type == inst.node_type
                               and i.status == RUNNING)
                    if live > t.min_workers:
                        inst.status = TERMINATING
                        try:
                            self.provider.terminate_node(inst.handle)
                        except Exception:
                            pass
                        inst.status = TERMINATED

    def _launch(self, launches: Dict[str, int]):
        for tname, count in launches.items():
            t = self.node_types[tname]
            for _ in range(count):
                inst = Instance(tname)
                self.instances.append(inst)
                inst.status = REQUESTED
                try:
                    inst.handle = self.provider.create_node(
                        t, {"ray_amd_instance": inst.id})
                except Exception:
                    inst.status = TERMINATED

    # ---------------- loop ----------------

    def _tick(self):
        nodes, demand = self._cluster_state()
        self._reconcile(nodes)
        now = time.time()
        if demand:
            if self._demand_since is None:
                self._demand_since = now
            if now - self._demand_since >= self.upscale_after_s:
                self._launch(self._plan(nodes, demand))
                self._demand_since = None
        else:
            self._demand_since = None
            # still honor min_workers
            launches = self._plan(nodes, [])
            if launches:
                self._launch(launches)

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="ray_amd_autoscaler_v2")
        self._thread.start()
        return self

    def _loop(self):
        while not self._stop.is_set():
            try:
                self._tick()
            except Exception:
                pass
            self._stop.wait(self.poll_s)

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(3)

    def summary(self) -> dict:
        out: Dict[str, Dict[str, int]] = {}
        for inst in self.instances:
            out.setdefault(inst.node_type, {}).setdefault(inst.status, 0)
            out[inst.node_type][inst.status] += 1
        return out
