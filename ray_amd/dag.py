"""Minimal DAG API (reference: python/ray/dag/). Full compiled-DAG
executor with overlap schedules is tracked for a later round; bind() /
execute() cover the basic composition surface."""
from __future__ import annotations


class DAGNode:
    def execute(self, *a, **k):
        raise NotImplementedError


class FunctionNode(DAGNode):
    def __init__(self, fn, args, kwargs):
        self._fn = fn
        self._args = args
        self._kwargs = kwargs

    def _resolve(self, v):
        if isinstance(v, DAGNode):
            return v.execute()
        return v

    def execute(self, *a, **k):
        args = [self._resolve(x) for x in self._args]
        kwargs = {k2: self._resolve(v) for k2, v in self._kwargs.items()}
        import ray_amd as ray

        args = [ray.get(x) if isinstance(x, ray.ObjectRef) else x for x in args]
        return self._fn.remote(*args, **kwargs)


class ClassNode(DAGNode):
    def __init__(self, cls, args, kwargs):
        self._cls = cls
        self._args = args
        self._kwargs = kwargs

    def execute(self, *a, **k):
        return self._cls.remote(*self._args, **self._kwargs)
