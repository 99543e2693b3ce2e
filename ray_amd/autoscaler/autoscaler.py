"""LocalAutoscaler: demand-driven node scaling on one machine.

Counterpart of the reference StandardAutoscaler loop
(autoscaler/_private/autoscaler.py) with the FakeMultiNode provider:
demand = queued lease counts reported by raylets (ray_syncer-style
resource reports carry a `pending` field) + explicit
sdk.request_resources; the provider starts/stops extra local raylets.
"""
from __future__ import annotations

import json
import threading
import time
from typing import Dict, List, Optional


class LocalAutoscaler:
    def __init__(
        self,
        cluster,  # cluster_utils.Cluster
        *,
        worker_resources: Optional[Dict[str, float]] = None,
        min_workers: int = 0,
        max_workers: int = 4,
        upscale_after_s: float = 0.5,
        idle_timeout_s: float = 60.0,
        poll_s: float = 0.25,
    ):
        self.cluster = cluster
        self.worker_resources = dict(worker_resources or {"CPU": 2})
        self.min_workers = min_workers
        self.max_workers = max_workers
        self.upscale_after_s = upscale_after_s
        self.idle_timeout_s = idle_timeout_s
        self.poll_s = poll_s
        self._workers: List = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._demand_since: Optional[float] = None
        self._idle_since: Dict[int, float] = {}

    def start(self):
        self._thread = threading.Thread(
            target=self._loop, name="ray_amd_autoscaler", daemon=True
        )
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread:
            self._thread.join(3)

    def _rt(self):
        from ray_amd._core import runtime as rtmod

        return rtmod.global_runtime()

    def _demand(self) -> int:
        """Queued lease requests across nodes + explicit requests."""
        rt = self._rt()
        table = rt.gcs_call("node_table", {})
        pending = sum(n.get("pending", 0) for n in table if n["alive"])
        try:
            v = rt.gcs_call(
                "kv_get", {"ns": "autoscaler", "key": b"request_resources"}
            )
            if v:
                req = json.loads(bytes(v).decode())
                want_cpu = req.get("num_cpus") or 0
                have_cpu = sum(
                    n["resources_total"].get("CPU", 0)
                    for n in table
                    if n["alive"]
                )
                if want_cpu > have_cpu:
                    pending += 1
        except Exception:
            pass
        return pending

    def _loop(self):
        while not self._stop.is_set():
            try:
                self._tick()
            except Exception:
                pass
            self._stop.wait(self.poll_s)

    def _tick(self):
        demand = self._demand()
        now = time.time()
        if demand > 0:
            if self._demand_since is None:
                self._demand_since = now
            if (
                now - self._demand_since >= self.upscale_after_s
                and len(self._workers) < self.max_workers
            ):
                h = self.cluster.add_node(resources=dict(self.worker_resources))
                self._workers.append(h)
                self._demand_since = None
        else:
            self._demand_since = None
        while len(self._workers) < self.min_workers:
            self._workers.append(
                self.cluster.add_node(resources=dict(self.worker_resources))
            )

    @property
    def num_workers(self) -> int:
        return len(self._workers)
