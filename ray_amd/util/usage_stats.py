"""Usage stats (reference: python/ray/_private/usage/usage_lib.py —
opt-out telemetry). This build has no egress: reports are written to
the session directory only (usage_stats.json), never transmitted;
the enable/disable surface and library-usage tagging match the
reference so operators can audit what WOULD be reported.
"""
from __future__ import annotations

import json
import os
import threading
import time
from typing import Set

_lock = threading.Lock()
_library_usages: Set[str] = set()
_extra_tags = {}

ENV_FLAG = "RAY_AMD_USAGE_STATS_ENABLED"
CONFIG_PATH = os.path.expanduser("~/.ray_amd/usage_stats_opt_out")


def usage_stats_enabled() -> bool:
    v = os.environ.get(ENV_FLAG)
    if v is not None:
        return v not in ("0", "false", "False")
    return not os.path.exists(CONFIG_PATH)


def set_usage_stats_enabled(enabled: bool) -> None:
    os.makedirs(os.path.dirname(CONFIG_PATH), exist_ok=True)
    if enabled:
        try:
            os.remove(CONFIG_PATH)
        except OSError:
            pass
    else:
        with open(CONFIG_PATH, "w") as f:
            f.write("opted out\n")


def record_library_usage(name: str) -> None:
    with _lock:
        _library_usages.add(name)


def record_extra_usage_tag(key: str, value: str) -> None:
    with _lock:
        _extra_tags[key] = value


def generate_report() -> dict:
    import platform

    try:
        import torch

        torch_v = torch.__version__
        n_gpus = torch.cuda.device_count() if torch.cuda.is_available() else 0
    except ImportError:
        torch_v, n_gpus = None, 0
    nodes = 0
    cpus = 0.0
    try:
        import ray_amd as ray

        if ray.is_initialized():
            alive = [n for n in ray.nodes() if n["Alive"]]
            nodes = len(alive)
            cpus = sum(n["Resources"].get("CPU", 0) for n in alive)
    except Exception:
        pass
    with _lock:
        libs = sorted(_library_usages)
        tags = dict(_extra_tags)
    return {
        "schema_version": "0.1",
        "timestamp": time.time(),
        "os": platform.system(),
        "python_version": platform.python_version(),
        "torch_version": torch_v,
        "num_nodes": nodes,
        "total_num_cpus": cpus,
        "total_num_gpus": n_gpus,
        "library_usages": libs,
        "extra_usage_tags": tags,
    }


def write_report(session_dir: str) -> str:
    """Called at shutdown when enabled; LOCAL file only (no egress)."""
    path = os.path.join(session_dir, "usage_stats.json")
    try:
        with open(path, "w") as f:
            json.dump(generate_report(), f, indent=1)
    except OSError:
        pass
    return path
