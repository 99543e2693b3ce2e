"""Dataset core (reference: python/ray/data/dataset.py — map :422,
map_batches :629, flat_map :1859, filter :2033, repartition :2269,
random_shuffle :2410, streaming_split :2629, groupby :3747, sort :4267,
iter_batches :6731, iter_torch_batches :6819; read_api.py:1370
read_parquet).

Blocks = pyarrow Tables in the object store; a Dataset is a lazy plan
over input blocks. Execution: per-block ray_amd tasks with a bounded
in-flight window for streaming consumption.
"""
from __future__ import annotations

import builtins
import itertools
import json
from typing import Any, Callable, Dict, Iterable, Iterator, List, Optional, Union

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq


class DataContext:
    _current = None

    def __init__(self):
        self.target_max_block_size = 128 * 1024 * 1024
        self.streaming_read_window = 8  # max in-flight blocks per consumer
        # streaming-executor resource budget (reference:
        # _internal/execution/streaming_executor.py resource manager):
        # stop launching new block tasks while the local object store is
        # above this fraction of capacity — in-flight blocks drain first
        self.object_store_memory_fraction = 0.8

    @classmethod
    def get_current(cls) -> "DataContext":
        if cls._current is None:
            cls._current = DataContext()
        return cls._current


def _ray():
    import ray_amd as ray

    return ray


def _store_pressure(ray) -> bool:
    """True while the local object store exceeds the streaming budget
    (cheap: cached 0.2 s)."""
    import time as _t

    now = _t.time()
    t0, val = getattr(_store_pressure, "_cache", (0.0, False))
    if now - t0 < 0.2:
        return val
    try:
        from ray_amd._core import runtime as _rtmod

        rt = _rtmod.global_runtime()
        r = rt._call_sync(rt.raylet.call("object_stats", {}), timeout=2)
        frac = DataContext.get_current().object_store_memory_fraction
        val = r["capacity"] > 0 and r["used"] / r["capacity"] > frac
    except Exception:
        val = False
    _store_pressure._cache = (now, val)
    return val


# ---------------- block helpers ----------------


def _to_table(block: Any) -> pa.Table:
    if isinstance(block, pa.Table):
        return block
    if isinstance(block, dict):
        arrays, fields = [], []
        for k, v in block.items():
            arr, shape = _np_to_arrow(v)
            meta = (
                {b"tensor_shape": json.dumps(shape).encode()} if shape else None
            )
            fields.append(pa.field(k, arr.type, metadata=meta))
            arrays.append(arr)
        return pa.Table.from_arrays(arrays, schema=pa.schema(fields))
    raise TypeError(f"bad block type {type(block)}")


def _np_to_arrow(v):
    """Returns (arrow array, inner_shape or None). Multi-dim tensors
    are stored as FixedSizeList with the inner shape kept in the field
    metadata (the role of the reference's ArrowTensorArray extension,
    data/extensions/)."""
    v = np.asarray(v)
    if v.ndim <= 1:
        return pa.array(v), None
    inner = list(v.shape[1:])
    return (
        pa.FixedSizeListArray.from_arrays(
            pa.array(np.ascontiguousarray(v).reshape(-1)), int(np.prod(inner))
        ),
        inner,
    )


def _col_to_numpy(col: pa.ChunkedArray, field: pa.Field = None) -> np.ndarray:
    t = col.type
    if pa.types.is_fixed_size_list(t):
        flat = col.combine_chunks().flatten().to_numpy(zero_copy_only=False)
        shape = None
        if field is not None and field.metadata and b"tensor_shape" in field.metadata:
            shape = json.loads(field.metadata[b"tensor_shape"])
        if shape:
            return flat.reshape((len(col), *shape))
        return flat.reshape(len(col), t.list_size)
    return col.to_numpy(zero_copy_only=False)


def _table_to_numpy(t: pa.Table) -> Dict[str, np.ndarray]:
    return {
        f.name: _col_to_numpy(t.column(f.name), f) for f in t.schema
    }


def _table_to_pandas(t: pa.Table):
    return t.to_pandas()


def _format_batch(t: pa.Table, fmt: Optional[str]):
    if fmt in (None, "default", "numpy"):
        return _table_to_numpy(t)
    if fmt == "pandas":
        return _table_to_pandas(t)
    if fmt in ("pyarrow", "arrow"):
        return t
    raise ValueError(f"unknown batch_format {fmt}")


def _batch_to_table(b) -> pa.Table:
    if isinstance(b, pa.Table):
        return b
    if isinstance(b, dict):
        return _to_table(b)
    try:
        import pandas as pd

        if isinstance(b, pd.DataFrame):
            return pa.Table.from_pandas(b, preserve_index=False)
    except ImportError:
        pass
    raise TypeError(f"map_batches fn returned unsupported type {type(b)}")


def _rows_of(t: pa.Table) -> List[dict]:
    cols = _table_to_numpy(t)
    names = list(cols)
    n = t.num_rows
    return [
        {name: cols[name][i] for name in names} for i in builtins.range(n)
    ]


# ---------------- plan ops (executed inside ray tasks) ----------------


def _op_map_batches(t: pa.Table, fn, fmt, batch_size) -> pa.Table:
    if batch_size is None or t.num_rows <= batch_size:
        out = fn(_format_batch(t, fmt))
        return _batch_to_table(out)
    parts = []
    for s in builtins.range(0, t.num_rows, batch_size):
        out = fn(_format_batch(t.slice(s, batch_size), fmt))
        parts.append(_batch_to_table(out))
    return pa.concat_tables(parts)


def _op_map(t: pa.Table, fn) -> pa.Table:
    rows = [fn(r) for r in _rows_of(t)]
    return _rows_to_table(rows)


def _op_flat_map(t: pa.Table, fn) -> pa.Table:
    rows = list(itertools.chain.from_iterable(fn(r) for r in _rows_of(t)))
    return _rows_to_table(rows)


def _op_filter(t: pa.Table, fn) -> pa.Table:
    mask = np.array([bool(fn(r)) for r in _rows_of(t)])
    return t.filter(pa.array(mask))


def _rows_to_table(rows: List[dict]) -> pa.Table:
    if not rows:
        return pa.table({})
    keys = rows[0].keys()
    return _to_table({k: np.asarray([r[k] for r in rows]) for k in keys})


class ActorPoolStrategy:
    """compute= strategy for map_batches: a fixed pool of UDF actors
    (reference: data ActorPoolMapOperator / ActorPoolStrategy)."""

    def __init__(self, size: Optional[int] = None,
                 min_size: Optional[int] = None,
                 max_size: Optional[int] = None):
        self.size = int(size or max_size or min_size or 2)


def _apply_ops(table_or_ref, ops: List[tuple]) -> pa.Table:
    t = table_or_ref
    for op in ops:
        kind = op[0]
        if kind == "map_batches":
            t = _op_map_batches(t, op[1], op[2], op[3])
        elif kind == "actor_map":
            # outside an actor pool (e.g. downstream task re-exec):
            # instantiate the UDF class per process as a fallback
            inst = op[1]() if isinstance(op[1], type) else op[1]
            t = _op_map_batches(t, inst, op[2], op[3])
        elif kind == "map":
            t = _op_map(t, op[1])
        elif kind == "flat_map":
            t = _op_flat_map(t, op[1])
        elif kind == "filter":
            t = _op_filter(t, op[1])
        elif kind == "read":
            t = op[1](t)  # t is a read task descriptor
    return t


# ---------------- distributed shuffle (reference: hash_shuffle.py) ----------------


def _hash_partition(t: pa.Table, key: Optional[str], num_parts: int,
                    seed: int = 0) -> List[pa.Table]:
    """Split a block into num_parts sub-tables by key hash (or random)."""
    n = t.num_rows
    if n == 0:
        return [t] * num_parts
    if key is None:
        rng = np.random.default_rng(seed)
        part = rng.integers(0, num_parts, n)
    else:
        vals = _col_to_numpy(t.column(key))
        if vals.dtype.kind in "OUS":
            part = np.array([hash(str(v)) % num_parts for v in vals])
        else:
            part = (vals.astype(np.int64, copy=False) * 2654435761 % 2**31
                    ) % num_parts
    return [t.filter(pa.array(part == p)) for p in builtins.range(num_parts)]


def _shuffle_exchange(ray, block_refs: List, key: Optional[str],
                      num_parts: int, seed: int = 0) -> List:
    """Two-stage map/reduce exchange: every block is hash-partitioned in
    parallel tasks, then each output partition concatenates its slice of
    every block (the reference's shuffle_operators exchange)."""

    @ray.remote
    def _map(block, num_parts=num_parts, key=key, seed=seed):
        return tuple(_hash_partition(block, key, num_parts, seed))

    @ray.remote
    def _reduce(*parts):
        tables = [p for p in parts if p.num_rows > 0]
        return pa.concat_tables(tables) if tables else pa.table({})

    map_refs = [
        _map.options(num_returns=num_parts).remote(b) for b in block_refs
    ]
    out = []
    for p in builtins.range(num_parts):
        cols = [mr[p] if isinstance(mr, list) else mr for mr in map_refs]
        if num_parts == 1:
            cols = map_refs
        out.append(_reduce.remote(*cols))
    return out


# ---------------- Dataset ----------------


class Dataset:
    def __init__(self, inputs: List[Any], ops: Optional[List[tuple]] = None,
                 owner=None):
        # inputs: list of ObjectRefs to pa.Table OR ("readtask", fn, arg)
        self._inputs = inputs
        self._ops = ops or []
        self._materialized: Optional[List[Any]] = None
        self._task_opts: Optional[dict] = None
        self._concurrency: Optional[int] = None  # per-op in-flight cap
        self._actor_pool_size: Optional[int] = None

    # ----- plan builders -----

    def _with_op(self, op: tuple) -> "Dataset":
        ds = Dataset(self._inputs, self._ops + [op])
        ds._task_opts = self._task_opts
        ds._concurrency = self._concurrency
        ds._actor_pool_size = self._actor_pool_size
        return ds

    def map(self, fn: Callable[[dict], dict], **kwargs) -> "Dataset":
        return self._with_op(("map", fn))

    def map_batches(
        self,
        fn: Callable,
        *,
        batch_size: Optional[int] = None,
        batch_format: Optional[str] = "numpy",
        compute=None,
        concurrency=None,
        fn_args=None,
        fn_kwargs=None,
        num_gpus: float = 0,
        **kwargs,
    ) -> "Dataset":
        if fn_args or fn_kwargs:
            base = fn
            a = tuple(fn_args or ())
            kw = dict(fn_kwargs or {})
            fn = lambda b: base(b, *a, **kw)  # noqa: E731
        if isinstance(fn, type):
            # class UDF -> real actor pool (reference:
            # ActorPoolMapOperator): one instance per pool actor
            if isinstance(compute, ActorPoolStrategy):
                size = compute.size
            elif isinstance(concurrency, int) and concurrency > 0:
                size = concurrency
            elif isinstance(concurrency, (tuple, list)):
                size = int(concurrency[-1])
            else:
                size = 2
            ds = self._with_op(("actor_map", fn, batch_format, batch_size))
            ds._actor_pool_size = max(1, size)
            return ds

        op = ("map_batches", fn, batch_format, batch_size)
        if num_gpus or kwargs.get("num_cpus") or concurrency:
            ds = self._with_op(op)
            if num_gpus or kwargs.get("num_cpus"):
                ds._task_opts = {
                    "num_gpus": num_gpus,
                    "num_cpus": kwargs.get("num_cpus", 1),
                }
            if concurrency:
                ds._concurrency = int(concurrency)
            return ds
        return self._with_op(op)

    def flat_map(self, fn) -> "Dataset":
        return self._with_op(("flat_map", fn))

    def filter(self, fn=None, *, expr=None) -> "Dataset":
        from .expressions import Expr

        if fn is None and isinstance(expr, Expr):
            # vectorized Expr filter (reference: data/expressions.py):
            # evaluate the mask per batch, not a python UDF per row
            def keep(batch, _e=expr):
                mask = np.asarray(_e.eval(batch), dtype=bool)
                return {k: np.asarray(v)[mask] for k, v in batch.items()}

            return self._with_op(("map_batches", keep, "numpy", None))
        if fn is None and expr is not None:
            code = compile(expr, "<filter_expr>", "eval")

            def fn(row, _code=code):
                return eval(_code, {}, dict(row))

        return self._with_op(("filter", fn))

    def add_column(self, name: str, fn) -> "Dataset":
        from .expressions import Expr

        def add(batch):
            batch = dict(batch)
            batch[name] = fn.eval(batch) if isinstance(fn, Expr) \
                else fn(batch)
            return batch

        return self._with_op(("map_batches", add, "numpy", None))

    def with_column(self, name: str, expr) -> "Dataset":
        """Expression-based column (reference: Dataset.with_columns +
        data/expressions.py)."""
        return self.add_column(name, expr)

    def with_columns(self, exprs: Dict[str, "Any"]) -> "Dataset":
        ds = self
        for name, e in exprs.items():
            ds = ds.add_column(name, e)
        return ds

    def drop_columns(self, cols: List[str]) -> "Dataset":
        def drop(t: pa.Table):
            return t.drop_columns([c for c in cols if c in t.column_names])

        return self._with_op(("map_batches", drop, "pyarrow", None))

    def select_columns(self, cols: List[str]) -> "Dataset":
        def sel(t: pa.Table):
            return t.select(cols)

        return self._with_op(("map_batches", sel, "pyarrow", None))

    def rename_columns(self, mapping: Dict[str, str]) -> "Dataset":
        def ren(t: pa.Table):
            return t.rename_columns(
                [mapping.get(c, c) for c in t.column_names]
            )

        return self._with_op(("map_batches", ren, "pyarrow", None))

    # ----- execution -----

    def _materialize_refs(self) -> List[Any]:
        """Run the plan; returns ObjectRefs of result blocks."""
        if self._materialized is not None:
            return self._materialized
        ray = _ray()
        ops = self._ops

        @ray.remote
        def _exec_block(block, ops=ops):
            t = block() if callable(block) else block
            return _apply_ops(t, ops)

        task = (
            _exec_block.options(**self._task_opts)
            if self._task_opts
            else _exec_block
        )
        refs = []
        for inp in self._inputs:
            refs.append(task.remote(inp))
        self._materialized = refs
        return refs

    def materialize(self) -> "Dataset":
        refs = self._materialize_refs()
        ds = Dataset(refs, [])
        ds._materialized = refs
        return ds

    def _iter_block_refs(self) -> Iterator[Any]:
        """Streaming execution through the operator-graph executor
        (data/_executor.py): fused task segments / actor-pool segments
        with per-op in-flight budgets, store-pressure backpressure on
        the source, and pool autoscaling."""
        ray = _ray()
        if self._materialized is not None:
            yield from self._materialized
            return
        window = DataContext.get_current().streaming_read_window
        from ._executor import build_chain

        ex = build_chain(
            ray, self._inputs, self._ops, _apply_ops,
            window=window,
            task_opts=self._task_opts,
            concurrency=self._concurrency,
            actor_pool_size=self._actor_pool_size,
            store_pressure=lambda: _store_pressure(ray),
        )
        yield from ex.run()

    def _iter_actor_pool(self, ops, window):
        """Blocks flow through a fixed pool of UDF actors (stateful,
        init-once); ordered results, windowed in-flight."""
        ray = _ray()

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
                return _apply_ops(t, self._ops)

        import builtins

        size = self._actor_pool_size or 2
        # NB: plain `range` is shadowed by the Dataset range() API here
        workers = [_PoolWorker.remote(ops) for _ in builtins.range(size)]
        try:
            pending = []
            i = 0
            inputs = iter(self._inputs)
            for inp in itertools.islice(inputs, max(window, size)):
                pending.append(workers[i % size].process.remote(inp))
                i += 1
            while pending:
                ref = pending.pop(0)
                nxt = next(inputs, None)
                if nxt is not None:
                    pending.append(workers[i % size].process.remote(nxt))
                    i += 1
                yield ref
        finally:
            for w in workers:
                try:
                    ray.kill(w)
                except Exception:
                    pass

    def _iter_tables(self) -> Iterator[pa.Table]:
        ray = _ray()
        for ref in self._iter_block_refs():
            yield ray.get(ref)

    # ----- consumption -----

    def count(self) -> int:
        ray = _ray()

        @ray.remote
        def _count(t):
            return t.num_rows

        return sum(ray.get([_count.remote(r) for r in self._materialize_refs()]))

    def schema(self):
        for t in self._iter_tables():
            if t.num_rows or t.num_columns:
                return t.schema
        return None

    def columns(self) -> List[str]:
        s = self.schema()
        return list(s.names) if s else []

    def take(self, limit: int = 20) -> List[dict]:
        out = []
        for t in self._iter_tables():
            out.extend(_rows_of(t))
            if len(out) >= limit:
                return out[:limit]
        return out

    def take_all(self) -> List[dict]:
        out = []
        for t in self._iter_tables():
            out.extend(_rows_of(t))
        return out

    def take_batch(self, batch_size: int = 20, *, batch_format="numpy"):
        acc = []
        n = 0
        for t in self._iter_tables():
            acc.append(t)
            n += t.num_rows
            if n >= batch_size:
                break
        t = pa.concat_tables(acc).slice(0, batch_size)
        return _format_batch(t, batch_format)

    def show(self, limit: int = 20):
        for r in self.take(limit):
            print(r)

    def iter_rows(self) -> Iterator[dict]:
        for t in self._iter_tables():
            yield from _rows_of(t)

    def iter_batches(
        self,
        *,
        batch_size: Optional[int] = 256,
        batch_format: Optional[str] = "numpy",
        drop_last: bool = False,
        local_shuffle_buffer_size: Optional[int] = None,
        prefetch_batches: int = 1,
        **kwargs,
    ) -> Iterator[Any]:
        carry: Optional[pa.Table] = None
        rng = np.random.default_rng(0)
        for t in self._iter_tables():
            if carry is not None and carry.num_rows:
                t = pa.concat_tables([carry, t])
                carry = None
            if local_shuffle_buffer_size:
                idx = rng.permutation(t.num_rows)
                t = t.take(pa.array(idx))
            if batch_size is None:
                yield _format_batch(t, batch_format)
                continue
            off = 0
            while t.num_rows - off >= batch_size:
                yield _format_batch(t.slice(off, batch_size), batch_format)
                off += batch_size
            if off < t.num_rows:
                carry = t.slice(off)
        if carry is not None and carry.num_rows and not drop_last:
            yield _format_batch(carry, batch_format)

    def iter_torch_batches(
        self,
        *,
        batch_size: Optional[int] = 256,
        dtypes=None,
        device: Optional[str] = None,
        collate_fn=None,
        drop_last: bool = False,
        local_shuffle_buffer_size: Optional[int] = None,
        prefetch_batches: int = 1,
        **kwargs,
    ):
        """Batches as torch tensors on `device`. With a CUDA target the
        CPU-side work (format, pin) runs in a background thread
        `prefetch_batches` ahead and H2D copies are non_blocking on
        pinned memory, overlapping the consumer's compute (SURVEY §2.9
        #7: async H2D staging)."""
        import queue as _q
        import threading as _th

        import torch

        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        use_cuda = str(device).startswith("cuda") and torch.cuda.is_available()

        def to_cpu_tensors(batch):
            if collate_fn is not None:
                return ("collated", collate_fn(batch))
            out = {}
            for k, v in batch.items():
                t = torch.as_tensor(np.ascontiguousarray(v))
                if dtypes is not None:
                    dt = dtypes.get(k) if isinstance(dtypes, dict) else dtypes
                    if dt is not None:
                        t = t.to(dt)
                if use_cuda:
                    t = t.pin_memory()
                out[k] = t
            return ("dict", out)

        src = self.iter_batches(
            batch_size=batch_size,
            batch_format="numpy",
            drop_last=drop_last,
            local_shuffle_buffer_size=local_shuffle_buffer_size,
        )

        def move(item):
            kind, val = item
            if kind == "collated":
                return val
            return {
                k: (t.to(device, non_blocking=True) if use_cuda
                    else t.to(device))
                for k, t in val.items()
            }

        if prefetch_batches and prefetch_batches > 0:
            q: "_q.Queue" = _q.Queue(maxsize=prefetch_batches)
            DONE = object()

            def producer():
                try:
                    for b in src:
                        q.put(to_cpu_tensors(b))
                except BaseException as e:  # noqa
                    q.put(("error", e))
                finally:
                    q.put(DONE)

            _th.Thread(target=producer, daemon=True).start()
            while True:
                item = q.get()
                if item is DONE:
                    return
                if item[0] == "error":
                    raise item[1]
                yield move(item)
        else:
            for b in src:
                yield move(to_cpu_tensors(b))

    def to_pandas(self, limit: Optional[int] = None):
        tables = list(self._iter_tables())
        t = pa.concat_tables(tables) if tables else pa.table({})
        df = t.to_pandas()
        return df.head(limit) if limit else df

    def to_arrow_refs(self):
        return list(self._materialize_refs())

    # ----- structure ops -----

    def repartition(self, num_blocks: int, **kwargs) -> "Dataset":
        ray = _ray()
        tables = list(self._iter_tables())
        if not tables:
            return self
        t = pa.concat_tables(tables)
        n = t.num_rows
        per = max(1, (n + num_blocks - 1) // num_blocks)
        refs = [
            ray.put(t.slice(i * per, per))
            for i in builtins.range(min(num_blocks, (n + per - 1) // per))
        ]
        ds = Dataset(refs, [])
        ds._materialized = refs
        return ds

    def random_shuffle(self, *, seed: Optional[int] = None, **kwargs) -> "Dataset":
        ray = _ray()
        refs = self._materialize_refs()
        if len(refs) <= 1:
            tables = list(self._iter_tables())
            if not tables:
                return self
            t = pa.concat_tables(tables)
            rng = np.random.default_rng(seed)
            t = t.take(pa.array(rng.permutation(t.num_rows)))
            ref = ray.put(t)
            ds = Dataset([ref], [])
            ds._materialized = [ref]
            return ds
        # distributed: hash-exchange rows randomly, then permute locally
        parts = _shuffle_exchange(ray, refs, None, len(refs), seed or 0)

        @ray.remote
        def _perm(t, s):
            rng = np.random.default_rng(s)
            return t.take(pa.array(rng.permutation(t.num_rows)))

        out = [_perm.remote(p, (seed or 0) + i) for i, p in enumerate(parts)]
        ds = Dataset(out, [])
        ds._materialized = out
        return ds

    def randomize_block_order(self, *, seed=None) -> "Dataset":
        import random as _random

        refs = list(self._materialize_refs())
        _random.Random(seed).shuffle(refs)
        ds = Dataset(refs, [])
        ds._materialized = refs
        return ds

    def split(self, n: int, *, equal: bool = False, locality_hints=None
              ) -> List["Dataset"]:
        ray = _ray()
        tables = list(self._iter_tables())
        t = pa.concat_tables(tables) if tables else pa.table({})
        rows = t.num_rows
        per = rows // n
        rem = rows % n
        out = []
        off = 0
        for i in builtins.range(n):
            k = per + (0 if equal else (1 if i < rem else 0))
            if equal:
                k = per
            ref = ray.put(t.slice(off, k))
            off += k
            ds = Dataset([ref], [])
            ds._materialized = [ref]
            out.append(ds)
        return out

    def streaming_split(self, n: int, *, equal: bool = True,
                        locality_hints=None) -> List["Dataset"]:
        return self.split(n, equal=equal)

    def split_at_indices(self, indices: List[int]) -> List["Dataset"]:
        ray = _ray()
        tables = list(self._iter_tables())
        t = pa.concat_tables(tables) if tables else pa.table({})
        bounds = [0] + list(indices) + [t.num_rows]
        out = []
        for i in builtins.range(len(bounds) - 1):
            ref = ray.put(t.slice(bounds[i], bounds[i + 1] - bounds[i]))
            ds = Dataset([ref], [])
            ds._materialized = [ref]
            out.append(ds)
        return out

    def train_test_split(self, test_size: Union[int, float], *,
                         shuffle: bool = False, seed=None):
        ds = self.random_shuffle(seed=seed) if shuffle else self
        n = ds.count()
        k = int(n * test_size) if isinstance(test_size, float) else test_size
        train, test = ds.split_at_indices([n - k])
        return train, test

    def union(self, *others: "Dataset") -> "Dataset":
        refs = list(self._materialize_refs())
        for o in others:
            refs.extend(o._materialize_refs())
        ds = Dataset(refs, [])
        ds._materialized = refs
        return ds

    def limit(self, n: int) -> "Dataset":
        ray = _ray()
        acc = []
        got = 0
        for t in self._iter_tables():
            if got + t.num_rows > n:
                acc.append(t.slice(0, n - got))
                got = n
                break
            acc.append(t)
            got += t.num_rows
        t = pa.concat_tables(acc) if acc else pa.table({})
        ref = ray.put(t)
        ds = Dataset([ref], [])
        ds._materialized = [ref]
        return ds

    def zip(self, other: "Dataset") -> "Dataset":
        ray = _ray()
        t1 = pa.concat_tables(list(self._iter_tables()))
        t2 = pa.concat_tables(list(other._iter_tables()))
        for name in t2.column_names:
            col_name = name if name not in t1.column_names else name + "_1"
            t1 = t1.append_column(col_name, t2.column(name))
        ref = ray.put(t1)
        ds = Dataset([ref], [])
        ds._materialized = [ref]
        return ds

    # ----- sort / groupby -----

    def sort(self, key: Union[str, List[str]], descending: bool = False) -> "Dataset":
        ray = _ray()
        t = pa.concat_tables(list(self._iter_tables()))
        keys = [key] if isinstance(key, str) else key
        order = "descending" if descending else "ascending"
        t = t.sort_by([(k, order) for k in keys])
        ref = ray.put(t)
        ds = Dataset([ref], [])
        ds._materialized = [ref]
        return ds

    def groupby(self, key: str) -> "GroupedData":
        return GroupedData(self, key)

    def join(self, other: "Dataset", *, on: str, how: str = "inner",
             num_partitions: Optional[int] = None) -> "Dataset":
        """Hash join on a key column (reference: data join operator).
        Both sides are hash-partitioned on `on`; per-partition joins run
        as parallel tasks."""
        ray = _ray()
        left_refs = self._materialize_refs()
        right_refs = other._materialize_refs()
        P = num_partitions or max(len(left_refs), len(right_refs), 1)
        lp = _shuffle_exchange(ray, left_refs, on, P)
        rp = _shuffle_exchange(ray, right_refs, on, P)

        @ray.remote
        def _join(lt, rt, on=on, how=how):
            if lt.num_rows == 0 and rt.num_rows == 0:
                return pa.table({})
            return lt.join(rt, keys=on, join_type=how)

        out = [_join.remote(lp[i], rp[i]) for i in builtins.range(P)]
        ds = Dataset(out, [])
        ds._materialized = out
        return ds

    def sum(self, on: str):
        return self._agg("sum", on)

    def min(self, on: str):
        return self._agg("min", on)

    def max(self, on: str):
        return self._agg("max", on)

    def mean(self, on: str):
        return self._agg("mean", on)

    def std(self, on: str):
        # pushdown: per-block (n, sum, sumsq) partials, driver combine
        ray = _ray()
        refs = self._materialize_refs()

        @ray.remote
        def _part(t, on=on):
            v = _col_to_numpy(t.column(on)) if t.num_rows else np.array([])
            return (int(v.size), float(np.sum(v)), float(np.sum(v * v)))

        parts = ray.get([_part.remote(r) for r in refs])
        n = sum(p[0] for p in parts)
        s = sum(p[1] for p in parts)
        ss = sum(p[2] for p in parts)
        return float(np.sqrt(max(ss - s * s / n, 0.0) / (n - 1)))

    def _agg(self, how: str, on: str):
        """Global aggregate with pushdown (reference: AggregateFn map/
        combine in _internal/planner/exchange): each block reduces
        remotely; only scalars travel to the driver."""
        ray = _ray()
        refs = self._materialize_refs()
        if len(refs) == 1:
            t = pa.concat_tables(list(self._iter_tables()))
            vals = _col_to_numpy(t.column(on))
            return getattr(np, how)(vals).item()

        @ray.remote
        def _part(t, on=on, how=how):
            if t.num_rows == 0:
                return None
            v = _col_to_numpy(t.column(on))
            if how == "mean":
                return (float(np.sum(v)), int(v.size))
            return getattr(np, how)(v).item()

        parts = [p for p in ray.get([_part.remote(r) for r in refs])
                 if p is not None]
        if how == "mean":
            return sum(p[0] for p in parts) / sum(p[1] for p in parts)
        if how == "sum":
            return type(parts[0])(sum(parts))
        return (min if how == "min" else max)(parts)

    def unique(self, column: str):
        t = pa.concat_tables(list(self._iter_tables()))
        return list(pa.compute.unique(t.column(column)).to_pylist())

    # ----- writes -----

    def write_parquet(self, path: str, **kwargs):
        import os

        os.makedirs(path, exist_ok=True)
        for i, t in enumerate(self._iter_tables()):
            pq.write_table(t, os.path.join(path, f"part-{i:05d}.parquet"))

    def write_csv(self, path: str, **kwargs):
        import os

        import pyarrow.csv as pcsv

        os.makedirs(path, exist_ok=True)
        for i, t in enumerate(self._iter_tables()):
            pcsv.write_csv(t, os.path.join(path, f"part-{i:05d}.csv"))

    def write_json(self, path: str, **kwargs):
        import json
        import os

        os.makedirs(path, exist_ok=True)
        for i, t in enumerate(self._iter_tables()):
            with open(os.path.join(path, f"part-{i:05d}.json"), "w") as f:
                for r in _rows_of(t):
                    f.write(json.dumps({k: _jsonable(v) for k, v in r.items()}) + "\n")

    def write_datasink(self, datasink, **kwargs):
        """Run datasink.write(block) as tasks over all blocks, then
        on_write_complete(results) on the driver."""
        ray = _ray()

        @ray.remote
        def _w(t, sink=datasink):
            return sink.write(t)

        results = ray.get([_w.remote(r) for r in self._materialize_refs()])
        datasink.on_write_complete(results)
        return results

    def write_numpy(self, path: str, *, column: str, **kwargs):
        import os

        os.makedirs(path, exist_ok=True)
        for i, t in enumerate(self._iter_tables()):
            np.save(os.path.join(path, f"part-{i:05d}.npy"),
                    _col_to_numpy(t.column(column)))

    # ----- misc -----

    def num_blocks(self) -> int:
        return len(self._inputs)

    def size_bytes(self) -> int:
        return sum(t.nbytes for t in self._iter_tables())

    def stats(self) -> str:
        from ._executor import StreamingExecutor

        base = f"Dataset(num_blocks={self.num_blocks()}, ops={len(self._ops)})"
        rep = StreamingExecutor.stats_report()
        return base + ("\n" + rep if rep else "")

    def __repr__(self):
        return f"Dataset(num_blocks={self.num_blocks()})"


def _jsonable(v):
    if isinstance(v, np.generic):
        return v.item()
    if isinstance(v, np.ndarray):
        return v.tolist()
    if isinstance(v, bytes):
        return v.decode("utf-8", "replace")
    return v


class GroupedData:
    def __init__(self, ds: Dataset, key: str):
        self._ds = ds
        self._key = key

    def _grouped(self):
        t = pa.concat_tables(list(self._ds._iter_tables()))
        return t.group_by(self._key)

    def _agg_distributed(self, on: str, how: str) -> Optional[Dataset]:
        """Multi-block: per-block PARTIAL aggregate (map-side combine —
        reference: aggregate pushdown in _internal/planner/exchange),
        hash-exchange only the combined partials, re-aggregate per
        partition. Shuffle volume is O(distinct keys per block), not
        O(rows)."""
        ray = _ray()
        refs = self._ds._materialize_refs()
        if len(refs) <= 1:
            return None
        key = self._key

        @ray.remote
        def _partial(t, on=on, how=how, key=key):
            if t.num_rows == 0:
                return t
            if how == "mean":
                return t.group_by(key).aggregate(
                    [(on, "sum"), (on, "count")])
            return t.group_by(key).aggregate([(on, how)])

        partial_refs = [_partial.remote(r) for r in refs]
        parts = _shuffle_exchange(ray, partial_refs, key, len(refs))

        @ray.remote
        def _final(t, on=on, how=how, key=key):
            if t.num_rows == 0:
                return t
            if how == "mean":
                g = t.group_by(key).aggregate(
                    [(f"{on}_sum", "sum"), (f"{on}_count", "sum")])
                s = _col_to_numpy(g.column(f"{on}_sum_sum"))
                n = _col_to_numpy(g.column(f"{on}_count_sum"))
                return pa.table({key: g.column(key),
                                 f"mean({on})": pa.array(s / n)})
            # sum of partial sums; min of mins; max of maxes
            refn = "sum" if how == "sum" else how
            g = t.group_by(key).aggregate([(f"{on}_{how}", refn)])
            return g.rename_columns(
                [f"{how}({on})" if c == f"{on}_{how}_{refn}" else c
                 for c in g.column_names])

        out_refs = [_final.remote(p) for p in parts]
        ds = Dataset(out_refs, [])
        ds._materialized = out_refs
        return ds

    def _wrap(self, t: pa.Table) -> Dataset:
        ray = _ray()
        ref = ray.put(t)
        ds = Dataset([ref], [])
        ds._materialized = [ref]
        return ds

    def count(self) -> Dataset:
        ray = _ray()
        refs = self._ds._materialize_refs()
        key = self._key
        if len(refs) > 1:
            # pushdown: per-block counts, shuffle partials, sum

            @ray.remote
            def _partial(t, key=key):
                if t.num_rows == 0:
                    return t
                return t.group_by(key).aggregate([(key, "count")])

            parts = _shuffle_exchange(
                ray, [_partial.remote(r) for r in refs], key, len(refs))

            @ray.remote
            def _final(t, key=key):
                if t.num_rows == 0:
                    return t
                g = t.group_by(key).aggregate([(f"{key}_count", "sum")])
                return g.rename_columns(
                    ["count()" if c == f"{key}_count_sum" else c
                     for c in g.column_names])

            out_refs = [_final.remote(p) for p in parts]
            ds = Dataset(out_refs, [])
            ds._materialized = out_refs
            return ds
        t = self._grouped().aggregate([(self._key, "count")])
        t = t.rename_columns([self._key, "count()"])
        return self._wrap(t)

    def sum(self, on: str) -> Dataset:
        d = self._agg_distributed(on, "sum")
        if d is not None:
            return d
        t = self._grouped().aggregate([(on, "sum")])
        t = t.rename_columns([f"sum({on})" if c == f"{on}_sum" else c
                              for c in t.column_names])
        return self._wrap(t)

    def mean(self, on: str) -> Dataset:
        d = self._agg_distributed(on, "mean")
        if d is not None:
            return d
        t = self._grouped().aggregate([(on, "mean")])
        t = t.rename_columns([f"mean({on})" if c == f"{on}_mean" else c
                              for c in t.column_names])
        return self._wrap(t)

    def min(self, on: str) -> Dataset:
        d = self._agg_distributed(on, "min")
        if d is not None:
            return d
        t = self._grouped().aggregate([(on, "min")])
        t = t.rename_columns([f"min({on})" if c == f"{on}_min" else c
                              for c in t.column_names])
        return self._wrap(t)

    def max(self, on: str) -> Dataset:
        d = self._agg_distributed(on, "max")
        if d is not None:
            return d
        t = self._grouped().aggregate([(on, "max")])
        t = t.rename_columns([f"max({on})" if c == f"{on}_max" else c
                              for c in t.column_names])
        return self._wrap(t)

    def map_groups(self, fn, *, batch_format="numpy") -> Dataset:
        t = pa.concat_tables(list(self._ds._iter_tables()))
        keys = _col_to_numpy(t.column(self._key))
        out = []
        for k in np.unique(keys):
            mask = pa.array(keys == k)
            sub = t.filter(mask)
            res = fn(_format_batch(sub, batch_format))
            out.append(_batch_to_table(res))
        return self._wrap(pa.concat_tables(out))


# ---------------- creation APIs ----------------


def from_items(items: List[Any], *, parallelism: int = -1) -> Dataset:
    ray = _ray()
    if items and isinstance(items[0], dict):
        rows = items
    else:
        rows = [{"item": it} for it in items]
    nb = parallelism if parallelism and parallelism > 0 else min(8, max(1, len(rows)))
    per = max(1, (len(rows) + nb - 1) // nb)
    refs = []
    for i in builtins.range(0, len(rows), per):
        refs.append(ray.put(_rows_to_table(rows[i : i + per])))
    ds = Dataset(refs, [])
    ds._materialized = refs
    return ds


def range(n: int, *, parallelism: int = -1, override_num_blocks=None) -> Dataset:
    ray = _ray()
    nb = override_num_blocks or (parallelism if parallelism > 0 else min(8, max(1, n)))
    per = max(1, (n + nb - 1) // nb)
    refs = []
    for s in builtins.range(0, n, per):
        e = min(s + per, n)
        refs.append(ray.put(pa.table({"id": np.arange(s, e, dtype=np.int64)})))
    ds = Dataset(refs, [])
    ds._materialized = refs
    return ds


def range_tensor(n: int, *, shape=(1,), parallelism: int = -1) -> Dataset:
    ds = range(n, parallelism=parallelism)

    def to_tensor(batch):
        ids = batch["id"]
        size = int(np.prod(shape))
        data = np.repeat(ids[:, None], size, axis=1).reshape((len(ids),) + tuple(shape))
        return {"data": data, "id": ids}

    return ds.map_batches(to_tensor)


def from_numpy(arr: np.ndarray, *, column: str = "data") -> Dataset:
    ray = _ray()
    ref = ray.put(_to_table({column: arr}))
    ds = Dataset([ref], [])
    ds._materialized = [ref]
    return ds


def from_pandas(df) -> Dataset:
    ray = _ray()
    ref = ray.put(pa.Table.from_pandas(df, preserve_index=False))
    ds = Dataset([ref], [])
    ds._materialized = [ref]
    return ds


def from_arrow(t: pa.Table) -> Dataset:
    ray = _ray()
    ref = ray.put(t)
    ds = Dataset([ref], [])
    ds._materialized = [ref]
    return ds


def _expand_paths(paths, suffix=None) -> List[str]:
    import glob as g
    import os

    if isinstance(paths, str):
        paths = [paths]
    out = []
    for p in paths:
        if os.path.isdir(p):
            out.extend(sorted(g.glob(os.path.join(p, "**", "*"), recursive=True)))
        else:
            out.extend(sorted(g.glob(p)) or [p])
    out = [p for p in out if os.path.isfile(p)]
    if suffix:
        matched = [p for p in out if p.endswith(suffix)]
        out = matched or out
    return out


def read_parquet(paths, *, columns=None, parallelism: int = -1, **kwargs) -> Dataset:
    ray = _ray()
    files = _expand_paths(paths, ".parquet")

    @ray.remote
    def _read(f, columns=columns):
        return pq.read_table(f, columns=columns)

    refs = [_read.remote(f) for f in files]
    ds = Dataset(refs, [])
    ds._materialized = refs
    return ds


def read_csv(paths, *, parallelism: int = -1, **kwargs) -> Dataset:
    ray = _ray()
    files = _expand_paths(paths, ".csv")

    @ray.remote
    def _read(f):
        import pyarrow.csv as pcsv

        return pcsv.read_csv(f)

    refs = [_read.remote(f) for f in files]
    ds = Dataset(refs, [])
    ds._materialized = refs
    return ds


def read_json(paths, *, parallelism: int = -1, **kwargs) -> Dataset:
    ray = _ray()
    files = _expand_paths(paths, ".json")

    @ray.remote
    def _read(f):
        import pyarrow.json as pjson

        return pjson.read_json(f)

    refs = [_read.remote(f) for f in files]
    ds = Dataset(refs, [])
    ds._materialized = refs
    return ds


def read_text(paths, *, encoding: str = "utf-8", drop_empty_lines=True,
              **kwargs) -> Dataset:
    """One row per line (reference: data/read_api.py read_text)."""
    ray = _ray()
    files = _expand_paths(paths)

    @ray.remote
    def _read(f, encoding=encoding, drop=drop_empty_lines):
        with open(f, "r", encoding=encoding) as fh:
            lines = fh.read().splitlines()
        if drop:
            lines = [l for l in lines if l.strip()]
        return pa.table({"text": pa.array(lines)})

    refs = [_read.remote(f) for f in files]
    ds = Dataset(refs, [])
    ds._materialized = refs
    return ds


def from_huggingface(hf_dataset, *, parallelism: int = -1) -> Dataset:
    """Zero-copy-ish ingestion of a HuggingFace `datasets.Dataset`
    (reference: data/read_api.py from_huggingface) — the HF dataset's
    arrow table is sliced into blocks and put into the object store."""
    ray = _ray()
    try:
        t = hf_dataset.data.table  # datasets.Dataset -> pyarrow Table
    except AttributeError:
        t = hf_dataset.with_format("arrow")[:]
    t = t.combine_chunks()
    n = t.num_rows
    if parallelism <= 0:
        parallelism = max(1, min(64, n // 10_000 or 1))
    step = (n + parallelism - 1) // parallelism
    import builtins

    refs = [
        ray.put(t.slice(i, min(step, n - i)))
        for i in builtins.range(0, n, step)  # module fn `range` shadows
    ]
    ds = Dataset(refs, [])
    ds._materialized = refs
    return ds


def read_binary_files(paths, *, include_paths: bool = False, **kwargs) -> Dataset:
    ray = _ray()
    files = _expand_paths(paths)

    @ray.remote
    def _read(f, include_paths=include_paths):
        with open(f, "rb") as fh:
            data = fh.read()
        cols = {"bytes": pa.array([data], type=pa.binary())}
        if include_paths:
            cols["path"] = pa.array([f])
        return pa.table(cols)

    refs = [_read.remote(f) for f in files]
    ds = Dataset(refs, [])
    ds._materialized = refs
    return ds


def read_numpy(paths, *, parallelism: int = -1, column: str = "data",
               **kwargs) -> Dataset:
    """.npy / .npz files, one block per file (reference:
    data/_internal/datasource numpy datasource). Lazy: files decode
    inside read tasks paced by the streaming executor."""
    files = _expand_paths(paths, ".npy") or _expand_paths(paths, ".npz")

    def _mk(f):
        def _read(f=f):
            arr = np.load(f, allow_pickle=False)
            if hasattr(arr, "files"):  # npz: one column per array
                return _to_table({k: arr[k] for k in arr.files})
            return _to_table({column: arr})

        return _read

    return Dataset([_mk(f) for f in files], [])


def read_webdataset(paths, *, parallelism: int = -1, **kwargs) -> Dataset:
    """WebDataset tar shards: files grouped by key prefix, one row per
    sample with a column per extension (reference:
    data/_internal/datasource webdataset). Pure tarfile, lazy blocks."""
    files = _expand_paths(paths, ".tar")

    def _mk(f):
        def _read(f=f):
            import tarfile

            samples = {}
            with tarfile.open(f) as tf:
                for m in tf.getmembers():
                    if not m.isfile():
                        continue
                    base = m.name
                    key, _, ext = base.partition(".")
                    samples.setdefault(key, {"__key__": key})[ext] = (
                        tf.extractfile(m).read()
                    )
            rows = sorted(samples.values(), key=lambda r: r["__key__"])
            cols = sorted({c for r in rows for c in r})
            return pa.table({c: [r.get(c) for r in rows] for c in cols})

        return _read

    return Dataset([_mk(f) for f in files], [])


class Datasource:
    """Plugin ABC for custom sources (reference:
    data/datasource/datasource.py). Implement get_read_tasks() to
    return zero-arg callables, each producing one pyarrow.Table
    block."""

    def get_read_tasks(self, parallelism: int):
        raise NotImplementedError

    def estimate_inmemory_data_size(self):
        return None


class Datasink:
    """Plugin ABC for custom sinks (reference:
    data/datasource/datasink.py). write() is called once per block
    inside a task; on_write_complete() once on the driver."""

    def write(self, block) -> Any:
        raise NotImplementedError

    def on_write_complete(self, results) -> None:
        pass


def read_datasource(datasource: Datasource, *, parallelism: int = -1,
                    **kwargs) -> Dataset:
    tasks = list(datasource.get_read_tasks(
        parallelism if parallelism > 0 else 8
    ))
    return Dataset(list(tasks), [])
