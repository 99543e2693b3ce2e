"""Central runtime configuration table (reference:
src/ray/common/ray_config_def.h — the 257-flag RayConfig). Every
tunable reads `RAY_AMD_<NAME>` from the environment at first access;
`ray_amd._config.config` is the singleton the subsystems consult.

Usage:
    from ray_amd._config import config
    cap = config.lease_request_cap
"""
from __future__ import annotations

import os
from typing import Any, Dict


class _Flag:
    __slots__ = ("env", "default", "cast", "doc")

    def __init__(self, env: str, default, cast, doc: str):
        self.env = env
        self.default = default
        self.cast = cast
        self.doc = doc


def _bool(v: str) -> bool:
    return v.lower() in ("1", "true", "yes", "on")


_FLAGS: Dict[str, _Flag] = {
    # object store
    "object_store_memory": _Flag(
        "RAY_AMD_OBJECT_STORE_MEMORY", 16 * 2**30, int,
        "per-node shm object-store capacity in bytes"),
    "inline_max_bytes": _Flag(
        "RAY_AMD_INLINE_MAX", 100 * 1024, int,
        "objects at or below this size stay inline in the owner's "
        "memory store (reference: max_direct_call_object_size)"),
    "pull_chunk_bytes": _Flag(
        "RAY_AMD_PULL_CHUNK_BYTES", 8 * 1024 * 1024, int,
        "chunk size for raylet-to-raylet object pulls "
        "(reference: object_manager_default_chunk_size)"),
    # worker pool
    "worker_prestart": _Flag(
        "RAY_AMD_WORKER_PRESTART", 8, int,
        "max workers prestarted per raylet (clamped to node CPUs)"),
    "worker_cap_factor": _Flag(
        "RAY_AMD_WORKER_CAP_FACTOR", 2.0, float,
        "worker-pool hard cap = CPUs * this factor"),
    # task submission
    "lease_request_cap": _Flag(
        "RAY_AMD_LEASE_REQUEST_CAP", 16, int,
        "max in-flight lease requests per scheduling key "
        "(reference: LeaseRequestRateLimiter)"),
    "lease_idle_grace_s": _Flag(
        "RAY_AMD_LEASE_IDLE_GRACE_S", 0.05, float,
        "idle worker lease kept this long for reuse before returning"),
    "default_max_retries": _Flag(
        "RAY_AMD_DEFAULT_MAX_RETRIES", 3, int,
        "default task max_retries (also bounds lineage re-executions)"),
    # timeouts / fault tolerance
    "gcs_reconnect_timeout_s": _Flag(
        "RAY_AMD_GCS_RECONNECT_TIMEOUT_S", 30.0, float,
        "how long clients retry reaching a restarted GCS"),
    "resource_report_period_s": _Flag(
        "RAY_AMD_RESOURCE_REPORT_PERIOD_S", 0.2, float,
        "raylet resource reporter loop period"),
    "streaming_backpressure_items": _Flag(
        "RAY_AMD_STREAMING_BACKPRESSURE_ITEMS", 64, int,
        "a streaming generator pauses when this many yielded items are "
        "unconsumed (reference: generator_backpressure_num_objects)"),
    # scheduling
    "scheduler_top_k_fraction": _Flag(
        "RAY_AMD_SCHEDULER_TOP_K_FRACTION", 0.2, float,
        "hybrid policy picks randomly among the best k = "
        "max(absolute, fraction*n) feasible nodes "
        "(reference: scheduler_top_k_fraction)"),
    "scheduler_top_k_absolute": _Flag(
        "RAY_AMD_SCHEDULER_TOP_K_ABSOLUTE", 1, int,
        "lower bound for the hybrid policy's top-k candidate pool"),
    # GPU data plane
    "ddp_bucket_cap_mb": _Flag(
        "RAY_AMD_DDP_BUCKET_MB", 128, int,
        "DDP gradient bucket size; xGMI rings are per-link bound so "
        "large buckets amortize latency (see train/torch.py)"),
    # testing
    "testing_rpc_failure": _Flag(
        "RAY_AMD_TESTING_RPC_FAILURE", "", str,
        "chaos hook 'method:prob' — drop matching RPCs (protocol.py)"),
}


class RayConfig:
    """Lazy env-backed flag table; attribute access returns the typed
    value. `describe()` lists every flag with default and doc."""

    def __init__(self):
        self._cache: Dict[str, Any] = {}

    def __getattr__(self, name: str):
        if name.startswith("_"):
            raise AttributeError(name)
        try:
            flag = _FLAGS[name]
        except KeyError:
            raise AttributeError(f"unknown config flag {name!r}") from None
        if name not in self._cache:
            raw = os.environ.get(flag.env)
            if raw is None:
                self._cache[name] = flag.default
            else:
                cast = _bool if flag.cast is bool else flag.cast
                self._cache[name] = cast(raw)
        return self._cache[name]

    def reload(self):
        """Re-read every flag from the environment (tests)."""
        self._cache.clear()

    @staticmethod
    def describe() -> Dict[str, dict]:
        return {
            name: {"env": f.env, "default": f.default, "doc": f.doc}
            for name, f in _FLAGS.items()
        }


config = RayConfig()
