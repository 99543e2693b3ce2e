"""Distributed tracing spans (reference:
python/ray/util/tracing/tracing_helper.py:183-193 — inject/extract
span context so spans cross process boundaries).

OTel-shaped but dependency-free: a span is {trace_id, span_id,
parent_id, name, start, end, attrs}. The caller's context is injected
into task/actor specs; the executing worker opens a child span around
user code and reports it through the task-event pipeline, so
`ray_amd timeline`/get_trace_events() reconstruct cross-process traces.
Enable via ray_amd.init(...) then enable_tracing(), or
RAY_AMD_TRACING=1.
"""
from __future__ import annotations

import contextvars
import os
import time
import uuid
from typing import Optional

_enabled = os.environ.get("RAY_AMD_TRACING") not in (None, "", "0")
_current: contextvars.ContextVar = contextvars.ContextVar(
    "ray_amd_span", default=None
)


def enable_tracing() -> None:
    global _enabled
    _enabled = True
    os.environ["RAY_AMD_TRACING"] = "1"  # workers inherit


def tracing_enabled() -> bool:
    return _enabled or os.environ.get("RAY_AMD_TRACING") not in (
        None, "", "0"
    )


def _new_id() -> str:
    return uuid.uuid4().hex[:16]


class Span:
    __slots__ = ("trace_id", "span_id", "parent_id", "name", "start",
                 "end", "attrs")

    def __init__(self, name: str, parent: Optional["Span"] = None,
                 trace_id: Optional[str] = None,
                 parent_id: Optional[str] = None):
        self.name = name
        self.trace_id = (trace_id or (parent.trace_id if parent else None)
                         or _new_id())
        self.parent_id = parent_id or (parent.span_id if parent else None)
        self.span_id = _new_id()
        self.start = time.time()
        self.end = None
        self.attrs = {}

    def finish(self):
        self.end = time.time()
        _record(self)

    def to_dict(self) -> dict:
        return {
            "trace_id": self.trace_id, "span_id": self.span_id,
            "parent_id": self.parent_id, "name": self.name,
            "start": self.start, "end": self.end, "attrs": self.attrs,
        }


_local_spans = []


def _record(s: Span):
    _local_spans.append(s.to_dict())
    # ship through the task-event pipeline when inside a worker
    try:
        from ray_amd._core import runtime as rtmod

        rt = rtmod.global_runtime()
        rt.raylet.notify("report_task_events",
                         {"events": [], "spans": [s.to_dict()]})
    except Exception:
        pass


_fallback_parent: Optional[dict] = None  # set by the executing worker


class span:
    """Context manager opening a child span of the current context."""

    def __init__(self, name: str, **attrs):
        self._name = name
        self._attrs = attrs
        self._span = None
        self._token = None

    def __enter__(self) -> Span:
        parent = _current.get()
        if parent is None and _fallback_parent is not None:
            self._span = Span(self._name,
                              trace_id=_fallback_parent["trace_id"],
                              parent_id=_fallback_parent["span_id"])
            self._span.attrs.update(self._attrs)
            self._token = _current.set(self._span)
            return self._span
        self._span = Span(self._name, parent=parent)
        self._span.attrs.update(self._attrs)
        self._token = _current.set(self._span)
        return self._span

    def __exit__(self, *exc):
        _current.reset(self._token)
        self._span.finish()


def current_span_context() -> Optional[dict]:
    """Injected into task specs (reference :183 inject)."""
    if not tracing_enabled():
        return None
    s = _current.get()
    if s is None:
        return None
    return {"trace_id": s.trace_id, "span_id": s.span_id}


def activate_remote_context(ctx: Optional[dict], name: str):
    """Worker side (reference :193 extract): open the execution span
    as a child of the caller's span."""
    if not tracing_enabled():
        return None
    global _fallback_parent
    parent_trace = ctx.get("trace_id") if ctx else None
    parent_span = ctx.get("span_id") if ctx else None
    s = Span(name, trace_id=parent_trace, parent_id=parent_span)
    token = _current.set(s)
    _fallback_parent = {"trace_id": s.trace_id, "span_id": s.span_id}
    return (s, token)


def finish_remote_context(handle):
    global _fallback_parent
    if handle is None:
        return
    s, token = handle
    _fallback_parent = None
    _current.reset(token)
    s.finish()


def get_trace_events() -> list:
    """Cluster-wide spans gathered from the GCS task-event store."""
    try:
        from ray_amd._core import runtime as rtmod

        rt = rtmod.global_runtime()
        evs = rt.gcs_call("timeline_events", {}) or []
        spans = [e["span"] for e in evs if isinstance(e, dict)
                 and e.get("span")]
        return spans + list(_local_spans)
    except Exception:
        return list(_local_spans)
