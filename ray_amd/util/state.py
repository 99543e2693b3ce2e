"""State API (reference: python/ray/util/state/api.py — list_actors
:793, list_nodes :885, list_tasks :1020, list_objects :1065)."""
from __future__ import annotations

from typing import List, Optional



def _rt():
    from ray_amd._core import runtime as r

    return r.global_runtime()


def list_actors(filters=None, limit: int = 100, **kwargs) -> List[dict]:
    rows = _rt().gcs_call("list_actors", {})
    out = []
    for r in rows:
        row = {
            "actor_id": bytes(r["actor_id"]).hex(),
            "class_name": r.get("class_name"),
            "state": r["state"],
            "name": r.get("name"),
            "namespace": r.get("namespace"),
            "node_id": bytes(r["node_id"]).hex() if r.get("node_id") else None,
            "num_restarts": r.get("num_restarts", 0),
        }
        if _match(row, filters):
            out.append(row)
    return out[:limit]


def list_nodes(filters=None, limit: int = 100, **kwargs) -> List[dict]:
    import ray_amd as ray

    rows = [
        {
            "node_id": n["NodeID"],
            "state": "ALIVE" if n["Alive"] else "DEAD",
            "resources_total": n["Resources"],
            "labels": n.get("Labels", {}),
        }
        for n in ray.nodes()
    ]
    return [r for r in rows if _match(r, filters)][:limit]


def list_placement_groups(filters=None, limit: int = 100, **kwargs) -> List[dict]:
    from ray_amd.util.placement_group import placement_group_table

    rows = [
        {"placement_group_id": k, **v}
        for k, v in placement_group_table().items()
    ]
    return [r for r in rows if _match(r, filters)][:limit]


def list_tasks(filters=None, limit: int = 10000, **kwargs) -> List[dict]:
    evs = _rt().raylet_call("report_task_events", {"events": [], "fetch": True})
    rows = [
        {
            "task_id": e["task_id"],
            "name": e["name"],
            "state": "FINISHED" if e.get("ok") else "FAILED",
            "start_time_ms": e["start"] * 1000,
            "end_time_ms": e["end"] * 1000,
            "worker_pid": e["pid"],
        }
        for e in evs
    ]
    return [r for r in rows if _match(r, filters)][:limit]


def list_objects(filters=None, limit: int = 100, **kwargs) -> List[dict]:
    st = _rt().raylet_call("object_stats", {})
    return [
        {
            "summary": True,
            "num_objects_in_store": st["num_objects"],
            "store_used_bytes": st["used"],
            "store_capacity_bytes": st["capacity"],
        }
    ]


def summarize_tasks(**kwargs) -> dict:
    tasks = list_tasks()
    by_name = {}
    for t in tasks:
        e = by_name.setdefault(t["name"], {"count": 0, "failed": 0})
        e["count"] += 1
        if t["state"] == "FAILED":
            e["failed"] += 1
    return by_name


def summarize_actors(**kwargs) -> dict:
    """Counts by class and state (reference: state summary API)."""
    out = {}
    for a in list_actors():
        e = out.setdefault(a.get("class_name", "Actor"), {})
        e[a["state"]] = e.get(a["state"], 0) + 1
    return out


def summarize_objects(**kwargs) -> dict:
    rows = list_objects()
    return {
        "total_objects": sum(r["num_objects_in_store"] for r in rows),
        "total_store_bytes": sum(r["store_used_bytes"] for r in rows),
        "nodes": len(rows),
    }


def get_log(filename: str = None, node_id: str = None, tail: int = 1000,
            **kwargs):
    """Yield log lines from the session's log directory (reference:
    ray.util.state.get_log)."""
    import os

    rt = _rt()
    logdir = os.path.join(rt.session_dir, "logs")
    if filename is None:
        raise ValueError(
            f"filename required; available: {sorted(os.listdir(logdir))}"
        )
    path = os.path.join(logdir, filename)
    with open(path, "r", errors="replace") as f:
        lines = f.readlines()
    for line in lines[-tail:]:
        yield line.rstrip("\n")


def list_logs(node_id: str = None, **kwargs):
    import os

    rt = _rt()
    logdir = os.path.join(rt.session_dir, "logs")
    return {"worker_out": sorted(os.listdir(logdir))}


def _match(row: dict, filters) -> bool:
    if not filters:
        return True
    for f in filters:
        key, op, val = f
        have = row.get(key)
        if op in ("=", "=="):
            if str(have) != str(val):
                return False
        elif op == "!=":
            if str(have) == str(val):
                return False
    return True


def node_debug_state() -> dict:
    """Per-handler event-loop stats + queue gauges from the local
    raylet and the GCS (reference: event_stats.cc DebugString /
    `ray status -v`)."""
    from ray_amd._core import runtime as _rtmod

    rt = _rtmod.global_runtime()
    out = {}
    out["raylet"] = rt._call_sync(rt.raylet.call("debug_state", {}), 10)
    out["gcs"] = rt.gcs_call("debug_state", {})
    return out
