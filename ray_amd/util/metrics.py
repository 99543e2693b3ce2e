"""Metrics API (reference: python/ray/util/metrics.py Counter/Gauge/
Histogram → stats pipeline). ray_amd aggregates in-process and
publishes snapshots to the GCS KV ("metrics" namespace); export_text()
renders Prometheus exposition format."""
from __future__ import annotations

import bisect
import json
import threading
import time
from typing import Dict, List, Optional, Tuple

_registry: Dict[str, "Metric"] = {}
_lock = threading.Lock()
_last_publish = 0.0


class Metric:
    def __init__(self, name: str, description: str = "",
                 tag_keys: Optional[Tuple[str, ...]] = None):
        self._name = name
        self._description = description
        self._tag_keys = tuple(tag_keys or ())
        self._default_tags: Dict[str, str] = {}
        self._values: Dict[tuple, float] = {}
        with _lock:
            _registry[name] = self

    def set_default_tags(self, tags: Dict[str, str]):
        self._default_tags = dict(tags)
        return self

    def _key(self, tags):
        merged = dict(self._default_tags)
        if tags:
            merged.update(tags)
        return tuple(sorted(merged.items()))

    @property
    def info(self):
        return {
            "name": self._name,
            "description": self._description,
            "tag_keys": self._tag_keys,
            "default_tags": self._default_tags,
        }


class Counter(Metric):
    def inc(self, value: float = 1.0, tags: Optional[Dict[str, str]] = None):
        k = self._key(tags)
        with _lock:
            self._values[k] = self._values.get(k, 0.0) + value
        _maybe_publish()


class Gauge(Metric):
    def set(self, value: float, tags: Optional[Dict[str, str]] = None):
        with _lock:
            self._values[self._key(tags)] = value
        _maybe_publish()


class Histogram(Metric):
    def __init__(self, name, description="", boundaries: Optional[List[float]] = None,
                 tag_keys=None):
        super().__init__(name, description, tag_keys)
        self._boundaries = sorted(boundaries or [0.1, 1, 10, 100, 1000])
        self._counts: Dict[tuple, List[int]] = {}
        self._sums: Dict[tuple, float] = {}

    def observe(self, value: float, tags: Optional[Dict[str, str]] = None):
        k = self._key(tags)
        with _lock:
            c = self._counts.setdefault(k, [0] * (len(self._boundaries) + 1))
            c[bisect.bisect_left(self._boundaries, value)] += 1
            self._sums[k] = self._sums.get(k, 0.0) + value
        _maybe_publish()


def _maybe_publish():
    """Push a metrics snapshot to the GCS KV at most every 2s."""
    global _last_publish
    now = time.time()
    if now - _last_publish < 2.0:
        return
    _last_publish = now
    try:
        from ray_amd._core import runtime as rtmod

        if not rtmod.is_initialized():
            return
        rt = rtmod.global_runtime()
        snap = json.dumps(export_dict()).encode()
        import os

        async def _put():
            await rt.gcs.call(
                "kv_put",
                {"ns": "metrics", "key": f"proc_{os.getpid()}".encode(),
                 "value": snap},
            )

        rt._run(_put())
    except Exception:
        pass


def export_dict() -> dict:
    out = {}
    with _lock:
        for name, m in _registry.items():
            if isinstance(m, Histogram):
                out[name] = {
                    "type": "histogram",
                    "boundaries": m._boundaries,
                    "counts": {str(k): v for k, v in m._counts.items()},
                    "sums": {str(k): v for k, v in m._sums.items()},
                }
            else:
                out[name] = {
                    "type": type(m).__name__.lower(),
                    "values": {str(k): v for k, v in m._values.items()},
                }
    return out


def export_text() -> str:
    """Prometheus exposition format of this process's metrics."""
    lines = []
    with _lock:
        for name, m in _registry.items():
            pname = name.replace(".", "_").replace("-", "_")
            lines.append(f"# HELP {pname} {m._description}")
            if isinstance(m, Histogram):
                lines.append(f"# TYPE {pname} histogram")
                for k, counts in m._counts.items():
                    labels = ",".join(f'{a}="{b}"' for a, b in k)
                    cum = 0
                    for b, c in zip(m._boundaries, counts):
                        cum += c
                        le = f'le="{b}"'
                        lab = f"{{{labels},{le}}}" if labels else f"{{{le}}}"
                        lines.append(f"{pname}_bucket{lab} {cum}")
                    total = sum(counts)
                    lab = f"{{{labels}}}" if labels else ""
                    lines.append(f"{pname}_count{lab} {total}")
                    lines.append(f"{pname}_sum{lab} {m._sums.get(k, 0.0)}")
            else:
                kind = "counter" if isinstance(m, Counter) else "gauge"
                lines.append(f"# TYPE {pname} {kind}")
                for k, v in m._values.items():
                    labels = ",".join(f'{a}="{b}"' for a, b in k)
                    lab = f"{{{labels}}}" if labels else ""
                    lines.append(f"{pname}{lab} {v}")
    return "\n".join(lines) + "\n"
