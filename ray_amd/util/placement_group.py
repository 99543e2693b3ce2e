"""Placement groups (reference: python/ray/util/placement_group.py:133).

GCS-side scheduling uses 2PC reserve/commit on raylets
(gcs.py _schedule_pg ↔ raylet bundle handlers), mirroring the
reference's prepare/commit protocol (node_manager.h:621-626).
"""
from __future__ import annotations

from typing import Dict, List, Optional

from .._core import ids as _ids
from .._core import runtime as _rt


class PlacementGroup:
    def __init__(self, pg_id: bytes, bundles: List[Dict[str, float]] = None):
        self.id = pg_id
        self._bundles = bundles or []

    @property
    def bundle_specs(self):
        return self._bundles

    @property
    def bundle_count(self):
        return len(self._bundles)

    def ready(self):
        """Returns an ObjectRef resolving when the PG is scheduled."""
        from .. import api

        @api.remote
        def _pg_ready(pg_id: bytes):
            rt = _rt.global_runtime()
            r = rt.gcs_call("pg_wait_ready", {"pg_id": pg_id, "timeout": 120.0})
            if r["state"] != "CREATED":
                raise RuntimeError(f"placement group state: {r['state']}")
            return True

        return _pg_ready.remote(self.id)

    def wait(self, timeout_seconds: float = 30) -> bool:
        rt = _rt.global_runtime()
        r = rt.gcs_call(
            "pg_wait_ready", {"pg_id": self.id, "timeout": timeout_seconds}
        )
        return r["state"] == "CREATED"

    def __reduce__(self):
        return (PlacementGroup, (self.id, self._bundles))


def placement_group(
    bundles: List[Dict[str, float]],
    strategy: str = "PACK",
    name: str = "",
    lifetime: Optional[str] = None,
    _max_cpu_fraction_per_node: Optional[float] = None,
) -> PlacementGroup:
    if strategy not in ("PACK", "SPREAD", "STRICT_PACK", "STRICT_SPREAD"):
        raise ValueError(f"invalid placement strategy {strategy}")
    if not bundles:
        raise ValueError("bundles must be non-empty")
    for b in bundles:
        if not b or any(v < 0 for v in b.values()):
            raise ValueError(f"invalid bundle {b}")
    rt = _rt.global_runtime()
    pg_id = _ids.new_pg_id()
    rt.gcs_call(
        "create_pg",
        {"pg_id": pg_id, "bundles": bundles, "strategy": strategy, "name": name},
    )
    return PlacementGroup(pg_id, bundles)


def remove_placement_group(pg: PlacementGroup):
    _rt.global_runtime().gcs_call("remove_pg", {"pg_id": pg.id})


def placement_group_table() -> dict:
    rt = _rt.global_runtime()
    rows = rt.gcs_call("pg_table", {})
    return {
        bytes(r["pg_id"]).hex(): {
            "name": r["name"],
            "state": r["state"],
            "strategy": r["strategy"],
            "bundles": {i: b for i, b in enumerate(r["bundles"])},
            "bundle_nodes": [
                bytes(n).hex() if n else None for n in r["bundle_nodes"]
            ],
        }
        for r in rows
    }


def get_current_placement_group() -> Optional[PlacementGroup]:
    return None
