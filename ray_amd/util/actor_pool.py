"""ActorPool (reference: python/ray/util/actor_pool.py, 508 LoC)."""
from __future__ import annotations

from typing import Any, Callable, Iterable, List


class ActorPool:
    def __init__(self, actors: List[Any]):
        import ray_amd as ray

        self._ray = ray
        self._idle = list(actors)
        self._future_to_actor = {}
        self._index_to_future = {}
        self._next_task_index = 0
        self._next_return_index = 0
        self._pending_submits = []

    def map(self, fn: Callable, values: Iterable):
        for v in values:
            self.submit(fn, v)
        while self.has_next():
            yield self.get_next()

    def map_unordered(self, fn: Callable, values: Iterable):
        for v in values:
            self.submit(fn, v)
        while self.has_next():
            yield self.get_next_unordered()

    def submit(self, fn: Callable, value):
        if self._idle:
            actor = self._idle.pop()
            future = fn(actor, value)
            self._future_to_actor[future] = (self._next_task_index, actor)
            self._index_to_future[self._next_task_index] = future
            self._next_task_index += 1
        else:
            self._pending_submits.append((fn, value))

    def has_next(self) -> bool:
        return bool(self._future_to_actor) or bool(self._pending_submits)

    def get_next(self, timeout=None):
        if not self.has_next():
            raise StopIteration("no more results")
        idx = self._next_return_index
        self._next_return_index += 1
        fut = self._index_to_future.pop(idx)
        res = self._ray.get(fut, timeout=timeout)
        _, actor = self._future_to_actor.pop(fut)
        self._return_actor(actor)
        return res

    def get_next_unordered(self, timeout=None):
        if not self.has_next():
            raise StopIteration("no more results")
        ready, _ = self._ray.wait(
            list(self._future_to_actor), num_returns=1, timeout=timeout
        )
        if not ready:
            raise TimeoutError("timed out waiting for result")
        fut = ready[0]
        idx, actor = self._future_to_actor.pop(fut)
        self._index_to_future.pop(idx, None)
        self._return_actor(actor)
        return self._ray.get(fut)

    def _return_actor(self, actor):
        self._idle.append(actor)
        while self._pending_submits and self._idle:
            fn, v = self._pending_submits.pop(0)
            self.submit(fn, v)

    def has_free(self) -> bool:
        return bool(self._idle)

    def pop_idle(self):
        return self._idle.pop() if self._idle else None

    def push(self, actor):
        self._idle.append(actor)
        while self._pending_submits and self._idle:
            fn, v = self._pending_submits.pop(0)
            self.submit(fn, v)
