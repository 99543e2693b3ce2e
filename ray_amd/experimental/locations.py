"""Object locations + dynamic resources (reference:
python/ray/experimental/locations.py get_object_locations,
python/ray/experimental/dynamic_resources.py set_resource)."""
from __future__ import annotations

from typing import Dict, List, Optional

from .._core import runtime as _rtmod


def get_object_locations(obj_refs: List, timeout_ms: int = -1) -> Dict:
    """{ref: {"node_ids": [hex...], "object_size": int|None,
    "did_spill": False}} — inline objects report no node (they live in
    the owner's memory store)."""
    rt = _rtmod.global_runtime()
    out = {}
    for ref in obj_refs:
        info = {"node_ids": [], "object_size": None, "did_spill": False}
        ent = rt.memory_store.get(ref.id)
        if ent is None and ref.owner_addr != rt.addr:
            try:
                c = rt._call_sync(rt._conn(ref.owner_addr))
                r = rt._call_sync(c.call("locate_object", {"id": ref.id}))
            except Exception:
                r = None
            if r and not r.get("inline"):
                ent = ("store", r["node_addr"], r["size"])
            elif r:
                ent = ("val", None)
        if ent is not None:
            if ent[0] == "store":
                info["object_size"] = ent[2]
                info["node_ids"] = [_node_of(rt, ent[1])]
        out[ref] = info
    return out


def _node_of(rt, raylet_addr: str) -> str:
    table = rt.gcs_call("node_table", {})
    for n in table:
        if n["addr"] == raylet_addr:
            return bytes(n["node_id"]).hex()
    return raylet_addr


def set_resource(resource_name: str, capacity: float,
                 node_id: Optional[str] = None):
    """Dynamically set a custom resource's capacity on a node (this
    node by default); capacity 0 deletes it."""
    rt = _rtmod.global_runtime()
    addr = rt.raylet_addr
    if node_id is not None:
        for n in rt.gcs_call("node_table", {}):
            if bytes(n["node_id"]).hex() == node_id:
                addr = n["addr"]
                break
        else:
            raise ValueError(f"no node {node_id}")

    async def do():
        c = await rt._conn(addr)
        return await c.call(
            "set_resource", {"resource": resource_name, "capacity": capacity}
        )

    return rt._call_sync(do())
