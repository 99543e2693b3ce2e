"""Autoscaler SDK (reference: ray.autoscaler.sdk.request_resources)."""
from __future__ import annotations

import json
from typing import List, Optional


def request_resources(num_cpus: Optional[int] = None,
                      bundles: Optional[List[dict]] = None):
    """Record an explicit resource demand the autoscaler must satisfy."""
    from ray_amd._core import runtime as rtmod

    rt = rtmod.global_runtime()
    payload = {"num_cpus": num_cpus, "bundles": bundles or []}
    rt.gcs_call(
        "kv_put",
        {"ns": "autoscaler", "key": b"request_resources",
         "value": json.dumps(payload).encode()},
    )


def get_requested_resources() -> dict:
    from ray_amd._core import runtime as rtmod

    rt = rtmod.global_runtime()
    v = rt.gcs_call("kv_get", {"ns": "autoscaler", "key": b"request_resources"})
    if not v:
        return {"num_cpus": None, "bundles": []}
    return json.loads(bytes(v).decode())
