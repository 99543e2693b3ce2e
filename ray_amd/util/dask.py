"""Dask-on-Ray scheduler (reference: python/ray/util/dask/scheduler.py
ray_dask_get): executes a dask task graph with one ray task per graph
node, dependencies passed as ObjectRefs so the object store holds
intermediates and independent nodes run in parallel.

A dask graph is a plain dict {key: spec} where spec is either a
literal, a key reference, or a task tuple (callable, *args) whose args
may themselves be task tuples, key references, or (possibly nested)
lists of those — so the scheduler needs no import of dask itself.
`enable_dask_on_ray()` registers it as dask's default get when dask is
installed.
"""
from __future__ import annotations

from typing import Any, Dict, Hashable, List


def _ray():
    import ray_amd

    return ray_amd


class _Dep:
    """Placeholder for a dependency's position in the resolved-args
    list shipped alongside the task template."""

    __slots__ = ("i",)

    def __init__(self, i: int):
        self.i = i


def _istask(x) -> bool:
    return isinstance(x, tuple) and bool(x) and callable(x[0])


def _is_key(x, dsk) -> bool:
    try:
        return x in dsk
    except TypeError:
        return False


def _build_template(expr, dsk, deps: List[Hashable]):
    """Replace key references in a task expression with _Dep slots;
    record the referenced keys in order."""
    if _istask(expr):
        return (expr[0],) + tuple(
            _build_template(a, dsk, deps) for a in expr[1:])
    if isinstance(expr, list):
        return [_build_template(a, dsk, deps) for a in expr]
    if _is_key(expr, dsk):
        deps.append(expr)
        return _Dep(len(deps) - 1)
    return expr


def _execute_template(expr, resolved):
    if isinstance(expr, _Dep):
        return resolved[expr.i]
    if _istask(expr):
        func = expr[0]
        args = [_execute_template(a, resolved) for a in expr[1:]]
        return func(*args)
    if isinstance(expr, list):
        return [_execute_template(a, resolved) for a in expr]
    return expr


def _toposort(dsk: Dict, keys: List[Hashable]) -> List[Hashable]:
    order: List[Hashable] = []
    seen = set()

    def visit(k, stack):
        if k in seen:
            return
        if k in stack:
            raise ValueError(f"cycle in dask graph at {k!r}")
        stack.add(k)
        deps: List[Hashable] = []
        _build_template(dsk[k], dsk, deps)
        for d in deps:
            visit(d, stack)
        stack.discard(k)
        seen.add(k)
        order.append(k)

    for k in keys:
        visit(k, set())
    return order


def _flatten_keys(keys):
    if isinstance(keys, list):
        out = []
        for k in keys:
            out.extend(_flatten_keys(k))
        return out
    return [keys]


def ray_dask_get(dsk: Dict, keys, **kwargs) -> Any:
    """Drop-in dask scheduler: `dask.compute(x, scheduler=ray_dask_get)`.

    Each graph node becomes one ray task; its dependency values arrive
    as ObjectRefs resolved by the worker, so results never bounce
    through the driver and sibling nodes run concurrently.
    """
    ray = _ray()

    @ray.remote
    def _node(template, *resolved):
        return _execute_template(template, list(resolved))

    flat = _flatten_keys(keys)
    refs: Dict[Hashable, Any] = {}
    for k in _toposort(dsk, flat):
        deps: List[Hashable] = []
        template = _build_template(dsk[k], dsk, deps)
        if not deps and not _istask(dsk[k]):
            refs[k] = ray.put(dsk[k])
            continue
        refs[k] = _node.remote(template, *[refs[d] for d in deps])

    def pack(ks):
        if isinstance(ks, list):
            return [pack(k) for k in ks]
        return ray.get(refs[ks])

    return pack(keys)


def enable_dask_on_ray():
    """Set ray_dask_get as dask's default scheduler (reference:
    util/dask enable_dask_on_ray). Requires dask to be installed."""
    import dask

    dask.config.set(scheduler=ray_dask_get)
    return dask.config


def disable_dask_on_ray():
    import dask

    dask.config.set(scheduler=None)
