"""Streaming operator-graph executor for ray_amd.data.

Feature counterpart of the reference's StreamingExecutor
(data/_internal/execution/streaming_executor.py:107, scheduling loop
:500, select_operator_to_run streaming_executor_state.py:898) with the
ResourceManager's per-operator budgets (resource_manager.py) and
backpressure policies (execution/backpressure_policy/) — redesigned
small: a linear chain of operators (input -> fused task segments /
actor-pool segments), each with its own in-flight budget and output
buffer, driven by one scheduling loop that launches downstream-first
(drain before produce), gates the source on object-store pressure, and
autoscales actor pools from queue depth.

Blocks move between operators as ObjectRefs — a block's bytes never
visit the driver.
"""
from __future__ import annotations

import time
from collections import deque
from typing import Any, Callable, Dict, Iterator, List, Optional


class OpStats:
    __slots__ = ("name", "launched", "completed", "errors", "busy_s",
                 "first_launch", "last_done", "peak_in_flight", "actors")

    def __init__(self, name: str):
        self.name = name
        self.launched = 0
        self.completed = 0
        self.errors = 0
        self.busy_s = 0.0
        self.first_launch = None
        self.last_done = None
        self.peak_in_flight = 0
        self.actors = 0

    def row(self) -> str:
        wall = (
            (self.last_done or time.time()) - self.first_launch
            if self.first_launch
            else 0.0
        )
        extra = f" actors={self.actors}" if self.actors else ""
        return (
            f"{self.name}: {self.completed}/{self.launched} blocks, "
            f"wall {wall:.2f}s, peak_in_flight {self.peak_in_flight}{extra}"
        )


class _Op:
    """One operator: an input queue, bounded in-flight tasks, and an
    ordered output buffer."""

    def __init__(self, name: str, budget: int, out_budget: int):
        self.name = name
        self.budget = max(1, budget)
        self.out_budget = max(2, out_budget)
        self.inq: deque = deque()
        self.inflight: deque = deque()  # ordered (ref, t0)
        self.outq: deque = deque()
        self.done_input = False
        self.stats = OpStats(name)

    # -- interface --
    def can_launch(self) -> bool:
        return (
            bool(self.inq)
            and len(self.inflight) < self.budget
            and len(self.outq) + len(self.inflight) < self.out_budget
        )

    def launch_one(self, ray):
        raise NotImplementedError

    def harvest(self, ray, ready_ids: set):
        """Move completed head-of-line tasks to the output buffer
        (order-preserving)."""
        while self.inflight and self.inflight[0][0] in ready_ids:
            ref, t0 = self.inflight.popleft()
            self.outq.append(ref)
            self.stats.completed += 1
            self.stats.busy_s += time.time() - t0
            self.stats.last_done = time.time()

    def pending_refs(self) -> List[Any]:
        return [r for r, _ in self.inflight]

    def finished(self) -> bool:
        return self.done_input and not self.inq and not self.inflight

    def close(self, ray):
        pass


class MapOp(_Op):
    """Fused stateless segment executed as one task per block."""

    def __init__(self, name: str, seg_ops: list, task_opts: Optional[dict],
                 budget: int, out_budget: int, apply_ops: Callable):
        super().__init__(name, budget, out_budget)
        self._seg = seg_ops
        self._opts = task_opts
        self._apply = apply_ops
        self._task = None

    def _ensure_task(self, ray):
        if self._task is None:
            seg = self._seg
            apply_ops = self._apply

            @ray.remote
            def _exec_block(block, seg=seg):
                t = block() if callable(block) else block
                return apply_ops(t, seg)

            self._task = (
                _exec_block.options(**self._opts) if self._opts else _exec_block
            )

    def launch_one(self, ray):
        self._ensure_task(ray)
        item = self.inq.popleft()
        ref = self._task.remote(item)
        self.inflight.append((ref, time.time()))
        self.stats.launched += 1
        self.stats.peak_in_flight = max(self.stats.peak_in_flight,
                                        len(self.inflight))
        if self.stats.first_launch is None:
            self.stats.first_launch = time.time()


class ActorMapOp(_Op):
    """Stateful segment on an autoscaling actor pool (reference:
    ActorPoolMapOperator with executor-driven pool sizing). Scales up
    while the input queue stays deeper than 2x the pool and below
    max_size; scales down when idle."""

    def __init__(self, name: str, seg_ops: list, max_size: int,
                 budget: int, out_budget: int, apply_ops: Callable):
        super().__init__(name, budget, out_budget)
        self._seg = seg_ops
        self.min_size = 1
        self.max_size = max(1, max_size)
        self._apply = apply_ops
        self._actors: list = []           # [(actor, inflight_count)]
        self._cls = None
        self._ref_actor: Dict[Any, int] = {}
        self._idle_since: Optional[float] = None

    def _ensure_cls(self, ray):
        if self._cls is not None:
            return
        apply_ops = self._apply
        seg = self._seg

        @ray.remote
        class _PoolWorker:
            def __init__(self, ops):
                self._insts = {}
                self._ops = [
                    (
                        ("map_batches", self._inst(i, op[1]), op[2], op[3])
                        if op[0] == "actor_map"
                        else op
                    )
                    for i, op in enumerate(ops)
                ]

            def _inst(self, i, cls):
                if i not in self._insts:
                    self._insts[i] = cls() if isinstance(cls, type) else cls
                return self._insts[i]

            def process(self, block):
                t = block() if callable(block) else block
                return apply_ops(t, self._ops)

        self._cls = _PoolWorker

    def _scale(self, ray):
        self._ensure_cls(ray)
        n = len(self._actors)
        if n < self.min_size or (
            n < self.max_size and len(self.inq) > 2 * max(n, 1)
        ):
            self._actors.append([self._cls.remote(self._seg), 0])
            self.stats.actors = len(self._actors)

    def can_launch(self) -> bool:
        if not super().can_launch():
            return False
        return True

    def launch_one(self, ray):
        self._scale(ray)
        item = self.inq.popleft()
        # least-loaded actor
        ent = min(self._actors, key=lambda e: e[1])
        ref = ent[0].process.remote(item)
        ent[1] += 1
        self._ref_actor[ref] = id(ent)
        self.inflight.append((ref, time.time()))
        self.stats.launched += 1
        self.stats.peak_in_flight = max(self.stats.peak_in_flight,
                                        len(self.inflight))
        if self.stats.first_launch is None:
            self.stats.first_launch = time.time()

    def harvest(self, ray, ready_ids: set):
        before = len(self.outq)
        super().harvest(ray, ready_ids)
        for ref in list(self.outq)[before:]:
            ent_id = self._ref_actor.pop(ref, None)
            if ent_id is not None:
                for e in self._actors:
                    if id(e) == ent_id:
                        e[1] = max(0, e[1] - 1)
                        break
        # scale down after sustained idleness
        if not self.inq and not self.inflight and len(self._actors) > self.min_size:
            if self._idle_since is None:
                self._idle_since = time.time()
            elif time.time() - self._idle_since > 2.0:
                actor, _ = self._actors.pop()
                try:
                    ray.kill(actor)
                except Exception:
                    pass
                self.stats.actors = len(self._actors)
                self._idle_since = None
        else:
            self._idle_since = None

    def close(self, ray):
        for actor, _ in self._actors:
            try:
                ray.kill(actor)
            except Exception:
                pass
        self._actors.clear()


class StreamingExecutor:
    """Drives a linear operator chain; yields the sink op's block refs
    in order as they complete."""

    #: stats of the most recent run (module-global, surfaced by
    #: Dataset.stats())
    last_stats: Optional[List[OpStats]] = None

    def __init__(self, ray, inputs: Iterator[Any], ops: List[_Op],
                 store_pressure: Callable[[], bool]):
        self.ray = ray
        self.inputs = iter(inputs)
        self.ops = ops
        self.store_pressure = store_pressure

    def run(self) -> Iterator[Any]:
        ray = self.ray
        ops = self.ops
        sink = ops[-1]
        src = ops[0]
        input_done = False
        try:
            while True:
                # 1. feed the source from the input iterator (bounded)
                while (not input_done and len(src.inq) < src.budget):
                    try:
                        src.inq.append(next(self.inputs))
                    except StopIteration:
                        input_done = True
                        src.done_input = True
                if input_done:
                    src.done_input = True

                # 2. harvest completed tasks everywhere
                all_pending = []
                for op in ops:
                    all_pending.extend(op.pending_refs())
                if all_pending:
                    ready, _ = ray.wait(
                        all_pending,
                        num_returns=len(all_pending),
                        timeout=0.02,
                    )
                    ready_ids = set(ready)
                    for op in ops:
                        op.harvest(ray, ready_ids)

                # 3. move outputs downstream; mark done_input edges
                for a, b in zip(ops, ops[1:]):
                    while a.outq:
                        b.inq.append(a.outq.popleft())
                    if a.finished() and not a.outq:
                        b.done_input = True

                # 4. yield sink results
                while sink.outq:
                    yield sink.outq.popleft()

                # 5. launch, downstream-first (drain before produce);
                #    the SOURCE op is additionally gated on store
                #    pressure unless nothing is in flight anywhere
                launched = False
                for op in reversed(ops):
                    while op.can_launch():
                        if (
                            op is src
                            and all_pending
                            and self.store_pressure()
                        ):
                            break
                        op.launch_one(ray)
                        launched = True

                if sink.finished():
                    while sink.outq:
                        yield sink.outq.popleft()
                    break
                if not launched and not all_pending:
                    # nothing running and nothing launchable: avoid a
                    # hot spin while upstream state settles
                    time.sleep(0.002)
        finally:
            StreamingExecutor.last_stats = [op.stats for op in ops]
            for op in ops:
                op.close(ray)

    @staticmethod
    def stats_report() -> str:
        if not StreamingExecutor.last_stats:
            return ""
        return "\n".join(
            ["Streaming executor (last run):"]
            + ["  " + s.row() for s in StreamingExecutor.last_stats]
        )


def build_chain(ray, inputs: List[Any], plan_ops: List[tuple],
                apply_ops: Callable, *, window: int,
                task_opts: Optional[dict], concurrency: Optional[int],
                actor_pool_size: Optional[int],
                store_pressure: Callable[[], bool]) -> StreamingExecutor:
    """Split the flat op chain into fused task segments and actor-pool
    segments (consecutive ops of the same kind fuse into one operator,
    the reference's operator-fusion rule)."""
    segments: List[tuple] = []  # (kind, [ops])
    for op in plan_ops:
        kind = "actor" if op[0] == "actor_map" else "task"
        if segments and segments[-1][0] == kind:
            segments[-1][1].append(op)
        else:
            segments.append((kind, [op]))
    if not segments:
        segments = [("task", [])]

    budget = window if not concurrency else min(window, concurrency)
    ops: List[_Op] = []
    for i, (kind, seg) in enumerate(segments):
        name = f"{kind}_seg{i}[" + ",".join(o[0] for o in seg) + "]"
        if kind == "actor":
            ops.append(
                ActorMapOp(name, seg, actor_pool_size or 2,
                           budget, 2 * budget, apply_ops)
            )
        else:
            ops.append(
                MapOp(name, seg, task_opts if i == len(segments) - 1 or
                      len(segments) == 1 else None,
                      budget, 2 * budget, apply_ops)
            )
    return StreamingExecutor(ray, inputs, ops, store_pressure)
