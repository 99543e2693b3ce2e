"""Distributed Queue backed by an actor (reference: python/ray/util/queue.py)."""
from __future__ import annotations

import asyncio
from typing import Any, List, Optional


class Empty(Exception):
    pass


class Full(Exception):
    pass


class _QueueActor:
    def __init__(self, maxsize: int = 0):
        self.q = asyncio.Queue(maxsize=maxsize)

    async def put(self, item, timeout=None):
        try:
            await asyncio.wait_for(self.q.put(item), timeout)
            return True
        except asyncio.TimeoutError:
            return False

    async def get(self, timeout=None):
        try:
            return True, await asyncio.wait_for(self.q.get(), timeout)
        except asyncio.TimeoutError:
            return False, None

    def put_nowait(self, item):
        try:
            self.q.put_nowait(item)
            return True
        except asyncio.QueueFull:
            return False

    def get_nowait(self):
        try:
            return True, self.q.get_nowait()
        except asyncio.QueueEmpty:
            return False, None

    def qsize(self):
        return self.q.qsize()

    def empty(self):
        return self.q.empty()

    def full(self):
        return self.q.full()


class Queue:
    def __init__(self, maxsize: int = 0, actor_options: Optional[dict] = None):
        import ray_amd as ray

        self._ray = ray
        opts = dict(actor_options or {})
        opts.setdefault("num_cpus", 0.1)
        self.actor = ray.remote(_QueueActor).options(**opts).remote(maxsize)

    def put(self, item: Any, block: bool = True, timeout: Optional[float] = None):
        if not block:
            ok = self._ray.get(self.actor.put_nowait.remote(item))
            if not ok:
                raise Full()
            return
        ok = self._ray.get(self.actor.put.remote(item, timeout))
        if not ok:
            raise Full()

    def get(self, block: bool = True, timeout: Optional[float] = None) -> Any:
        if not block:
            ok, v = self._ray.get(self.actor.get_nowait.remote())
            if not ok:
                raise Empty()
            return v
        ok, v = self._ray.get(self.actor.get.remote(timeout))
        if not ok:
            raise Empty()
        return v

    def put_nowait(self, item):
        return self.put(item, block=False)

    def get_nowait(self):
        return self.get(block=False)

    def put_async(self, item):
        return self.actor.put.remote(item)

    def get_async(self):
        return self.actor.get.remote()

    def qsize(self) -> int:
        return self._ray.get(self.actor.qsize.remote())

    def size(self) -> int:
        return self.qsize()

    def empty(self) -> bool:
        return self._ray.get(self.actor.empty.remote())

    def full(self) -> bool:
        return self._ray.get(self.actor.full.remote())

    def put_batch(self, items: List[Any]):
        for i in items:
            self.put(i)

    def get_batch(self, n: int) -> List[Any]:
        return [self.get() for _ in range(n)]

    def shutdown(self, force: bool = False):
        self._ray.kill(self.actor)
