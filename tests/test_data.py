"""ray_amd.data tests (reference model: python/ray/data/tests/)."""
import numpy as np
import pytest

import ray_amd as ray
import ray_amd.data as rd


def test_range_count_take(ray_start_regular):
    ds = rd.range(100)
    assert ds.count() == 100
    rows = ds.take(5)
    assert [r["id"] for r in rows] == [0, 1, 2, 3, 4]


def test_from_items_map(ray_start_regular):
    ds = rd.from_items([{"x": i} for i in range(20)])
    out = ds.map(lambda r: {"y": r["x"] * 2}).take_all()
    assert sorted(r["y"] for r in out) == [2 * i for i in range(20)]


def test_map_batches_numpy(ray_start_regular):
    ds = rd.range(64).map_batches(lambda b: {"id": b["id"] * 10})
    assert sorted(r["id"] for r in ds.take_all()) == [10 * i for i in range(64)]


def test_filter_flat_map(ray_start_regular):
    ds = rd.range(10).filter(lambda r: r["id"] % 2 == 0)
    assert ds.count() == 5
    ds2 = rd.range(3).flat_map(lambda r: [{"v": r["id"]}, {"v": r["id"]}])
    assert ds2.count() == 6


def test_batch_iteration(ray_start_regular):
    ds = rd.range(100)
    batches = list(ds.iter_batches(batch_size=32))
    sizes = [len(b["id"]) for b in batches]
    assert sum(sizes) == 100
    assert sizes[:3] == [32, 32, 32]


def test_iter_torch_batches(ray_start_regular):
    import torch

    ds = rd.range(16)
    batches = list(ds.iter_torch_batches(batch_size=8, device="cpu"))
    assert len(batches) == 2
    assert isinstance(batches[0]["id"], torch.Tensor)


def test_repartition_split(ray_start_regular):
    ds = rd.range(100).repartition(4)
    assert ds.num_blocks() == 4
    parts = ds.split(2)
    assert parts[0].count() + parts[1].count() == 100


def test_random_shuffle_preserves_rows(ray_start_regular):
    ds = rd.range(50).random_shuffle(seed=42)
    vals = sorted(r["id"] for r in ds.take_all())
    assert vals == list(range(50))


def test_sort_groupby(ray_start_regular):
    ds = rd.from_items([{"k": i % 3, "v": i} for i in range(12)])
    s = ds.sort("v", descending=True).take(1)
    assert s[0]["v"] == 11
    agg = ds.groupby("k").sum("v").take_all()
    total = {int(r["k"]): r for r in agg}
    # groups: k=0 -> 0+3+6+9=18, k=1 -> 1+4+7+10=22, k=2 -> 2+5+8+11=26
    sums = sorted(float(list(r.values())[1]) for r in agg)
    assert sums == [18.0, 22.0, 26.0]


def test_aggregates(ray_start_regular):
    ds = rd.range(10)
    assert ds.sum("id") == 45
    assert ds.min("id") == 0
    assert ds.max("id") == 9
    assert ds.mean("id") == pytest.approx(4.5)


def test_limit_union_zip(ray_start_regular):
    a = rd.range(10).limit(3)
    assert a.count() == 3
    b = rd.range(5)
    u = a.union(b)
    assert u.count() == 8
    z = rd.range(4).zip(rd.range(4).map_batches(lambda x: {"other": x["id"] + 1}))
    rows = z.take_all()
    assert all(r["other"] == r["id"] + 1 for r in rows)


def test_parquet_roundtrip(ray_start_regular, tmp_path):
    ds = rd.range(30)
    path = str(tmp_path / "pq")
    ds.write_parquet(path)
    back = rd.read_parquet(path)
    assert back.count() == 30
    assert sorted(r["id"] for r in back.take_all()) == list(range(30))


def test_csv_roundtrip(ray_start_regular, tmp_path):
    ds = rd.from_items([{"a": i, "b": f"s{i}"} for i in range(10)])
    path = str(tmp_path / "csv")
    ds.write_csv(path)
    back = rd.read_csv(path)
    assert back.count() == 10


def test_tensor_columns(ray_start_regular):
    arr = np.random.rand(16, 4).astype(np.float32)
    ds = rd.from_numpy(arr)
    b = ds.take_batch(16)
    np.testing.assert_allclose(b["data"], arr, rtol=1e-6)


def test_preprocessor_standard_scaler(ray_start_regular):
    ds = rd.from_items([{"x": float(i)} for i in range(10)])
    from ray_amd.data import StandardScaler

    sc = StandardScaler(["x"]).fit(ds)
    out = sc.transform(ds)
    vals = np.array([r["x"] for r in out.take_all()])
    assert abs(vals.mean()) < 1e-6


def test_train_test_split(ray_start_regular):
    tr, te = rd.range(100).train_test_split(0.2)
    assert tr.count() == 80 and te.count() == 20


def test_dataset_shard_in_trainer(ray_start_regular, tmp_path):
    from ray_amd.train import FailureConfig, RunConfig, ScalingConfig
    from ray_amd.train.torch import TorchTrainer

    def loop(config):
        import ray_amd.train as train

        shard = train.get_dataset_shard("train")
        n = sum(len(b["id"]) for b in shard.iter_batches(batch_size=None))
        train.report({"rows": n})

    ds = rd.range(40)
    t = TorchTrainer(
        loop,
        scaling_config=ScalingConfig(num_workers=2),
        run_config=RunConfig(
            name="shard", storage_path=str(tmp_path),
            # retry transient worker-group failures under full-suite load
            failure_config=FailureConfig(max_failures=2),
        ),
        datasets={"train": ds},
    )
    res = t.fit()
    assert res.error is None, res.error
    assert res.metrics["rows"] == 20


def test_distributed_shuffle_exchange(ray_start_regular):
    # many blocks -> hash-exchange path
    ds = rd.range(200, parallelism=8)
    sh = ds.random_shuffle(seed=7)
    vals = sorted(r["id"] for r in sh.take_all())
    assert vals == list(range(200))
    assert sh.num_blocks() == 8


def test_distributed_groupby(ray_start_regular):
    ds = rd.from_items([{"k": i % 5, "v": i} for i in range(100)],
                       parallelism=8)
    agg = ds.groupby("k").sum("v").take_all()
    got = {int(r["k"]): float(r["sum(v)"]) for r in agg}
    expect = {}
    for i in range(100):
        expect[i % 5] = expect.get(i % 5, 0) + i
    assert got == expect


def test_aggregate_pushdown_multiblock(ray_start_regular):
    """Map-side combine: grouped mean/count and global aggregates on a
    multi-block dataset reduce per block before any shuffle/driver
    transfer (reference: AggregateFn map/combine pushdown)."""
    ds = rd.from_items([{"k": i % 4, "v": float(i)} for i in range(80)],
                       parallelism=8)
    mean = {int(r["k"]): r["mean(v)"]
            for r in ds.groupby("k").mean("v").take_all()}
    cnt = {int(r["k"]): r["count()"]
           for r in ds.groupby("k").count().take_all()}
    mn = {int(r["k"]): r["min(v)"]
          for r in ds.groupby("k").min("v").take_all()}
    import numpy as _np

    for k in range(4):
        vs = _np.array([float(i) for i in range(80) if i % 4 == k])
        assert abs(mean[k] - vs.mean()) < 1e-9
        assert cnt[k] == 20
        assert mn[k] == vs.min()
    # global aggregates on the same multi-block dataset
    assert ds.sum("v") == sum(range(80))
    assert abs(ds.mean("v") - 39.5) < 1e-9
    assert ds.min("v") == 0.0 and ds.max("v") == 79.0
    assert abs(ds.std("v") - _np.std(_np.arange(80.0), ddof=1)) < 1e-9


def test_hash_join(ray_start_regular):
    left = rd.from_items(
        [{"k": i, "a": i * 10} for i in range(20)], parallelism=4
    )
    right = rd.from_items(
        [{"k": i, "b": i * 100} for i in range(10, 30)], parallelism=4
    )
    joined = left.join(right, on="k").take_all()
    assert len(joined) == 10  # keys 10..19
    for r in joined:
        assert r["a"] == r["k"] * 10 and r["b"] == r["k"] * 100


def test_read_text_and_from_huggingface(ray_start_regular, tmp_path):
    from ray_amd import data as ray_data

    f = tmp_path / "lines.txt"
    f.write_text("alpha\n\nbeta\ngamma\n")
    ds = ray_data.read_text(str(f))
    assert [r["text"] for r in ds.take_all()] == ["alpha", "beta", "gamma"]

    import datasets as hf

    hfd = hf.Dataset.from_dict({"x": list(range(100)), "y": ["a"] * 100})
    ds2 = ray_data.from_huggingface(hfd)
    assert ds2.count() == 100
    out = ds2.map_batches(lambda b: {"x2": b["x"] * 2}).take(3)
    assert out[1]["x2"] == 2


def test_map_batches_concurrency_cap(ray_start_regular):
    from ray_amd import data as ray_data

    ds = ray_data.range(64, override_num_blocks=16).map_batches(
        lambda b: b, concurrency=2
    )
    assert ds.count() == 64  # capped window still processes everything


def test_map_batches_actor_pool(ray_start_regular, tmp_path):
    """Class UDFs run on a fixed actor pool: init once per actor,
    state persists across blocks (reference: ActorPoolMapOperator)."""
    from ray_amd import data as ray_data
    from ray_amd.data.dataset import ActorPoolStrategy

    init_dir = tmp_path

    class AddModel:
        def __init__(self):
            import os
            import time as _t

            with open(init_dir / f"init_{os.getpid()}_{_t.time_ns()}",
                      "w") as f:
                f.write("x")
            self.seen = 0

        def __call__(self, batch):
            self.seen += 1
            return {"v": batch["id"] + 1000}

    ds = ray_data.range(80, override_num_blocks=8).map_batches(
        AddModel, compute=ActorPoolStrategy(size=3)
    )
    out = ds.take_all()
    assert sorted(r["v"] for r in out) == list(range(1000, 1080))
    inits = list(init_dir.glob("init_*"))
    assert len(inits) == 3  # one instance per pool actor, not per block


def test_streaming_executor_budgets_and_stats(ray_start_regular):
    """Operator-graph executor: per-op in-flight stays under budget,
    mixed task/actor chains split into segments, op-level stats
    reported (reference: streaming_executor.py + resource_manager)."""
    from ray_amd.data import DataContext
    from ray_amd.data._executor import StreamingExecutor

    ctx = DataContext.get_current()
    prev = ctx.streaming_read_window
    ctx.streaming_read_window = 3
    try:
        class AddOne:
            def __call__(self, b):
                return {"id": b["id"] + 1}

        ds = (
            rd.range(40, override_num_blocks=10)
            .map_batches(lambda b: {"id": b["id"] * 2})
            .map_batches(AddOne, concurrency=2)
            .map_batches(lambda b: {"id": b["id"] + 100})
        )
        rows = sorted(r["id"] for r in ds.take_all())
        assert rows == sorted(2 * i + 101 for i in range(40))
        stats = StreamingExecutor.last_stats
        assert stats is not None and len(stats) == 3  # task, actor, task
        for s in stats:
            assert s.completed == 10
            assert s.peak_in_flight <= 3
        actor_stats = stats[1]
        assert actor_stats.actors >= 1
        report = ds.stats()
        assert "Streaming executor" in report
    finally:
        ctx.streaming_read_window = prev


def test_streaming_executor_order_preserved(ray_start_regular):
    ds = rd.range(100, override_num_blocks=20).map_batches(
        lambda b: {"id": b["id"]}
    )
    got = [r["id"] for r in ds.take_all()]
    assert got == list(range(100))


def test_read_numpy_and_npz(ray_start_regular, tmp_path):
    a = np.arange(20, dtype=np.float32)
    np.save(tmp_path / "a.npy", a)
    ds = rd.read_numpy(str(tmp_path / "a.npy"))
    rows = ds.take_all()
    assert [float(r["data"]) for r in rows] == list(map(float, a))
    np.savez(tmp_path / "b.npz", x=np.arange(4), y=np.arange(4) * 2)
    ds2 = rd.read_numpy(str(tmp_path / "b.npz"))
    rows = ds2.take_all()
    assert [r["y"] for r in rows] == [0, 2, 4, 6]


def test_read_webdataset(ray_start_regular, tmp_path):
    import io
    import tarfile

    tarp = tmp_path / "shard-000.tar"
    with tarfile.open(tarp, "w") as tf:
        for key in ("s0", "s1"):
            for ext, payload in (("txt", f"text-{key}".encode()),
                                 ("cls", b"7")):
                data = io.BytesIO(payload)
                info = tarfile.TarInfo(f"{key}.{ext}")
                info.size = len(payload)
                tf.addfile(info, data)
    ds = rd.read_webdataset(str(tarp))
    rows = ds.take_all()
    assert len(rows) == 2
    assert rows[0]["__key__"] == "s0" and rows[0]["txt"] == b"text-s0"
    assert rows[1]["cls"] == b"7"


def test_datasource_datasink_plugins(ray_start_regular):
    class SquaresSource(rd.Datasource):
        def get_read_tasks(self, parallelism):
            def mk(i):
                def read(i=i):
                    import pyarrow as pa

                    return pa.table({"x": [i * i]})

                return read

            return [mk(i) for i in range(5)]

    ds = rd.read_datasource(SquaresSource())
    assert sorted(r["x"] for r in ds.take_all()) == [0, 1, 4, 9, 16]

    collected = []

    class ListSink(rd.Datasink):
        def write(self, block):
            return block.num_rows

        def on_write_complete(self, results):
            collected.extend(results)

    ds.write_datasink(ListSink())
    assert sum(collected) == 5


def test_expressions_filter_and_with_columns(ray_start_regular):
    """Expression API (reference: data/expressions.py col()/lit()):
    vectorized filter masks and computed columns."""
    from ray_amd.data.expressions import col, lit

    ds = rd.from_items([{"x": i, "y": i * 2} for i in range(20)],
                       parallelism=4)
    out = ds.filter(expr=(col("x") > 5) & (col("y") < 30)).take_all()
    assert [r["x"] for r in out] == [6, 7, 8, 9, 10, 11, 12, 13, 14]

    ds2 = ds.with_columns({
        "z": col("x") + col("y") * lit(10),
        "neg": -col("x"),
    })
    rows = ds2.take(3)
    assert rows[1]["z"] == 1 + 2 * 10 and rows[1]["neg"] == -1

    # is_in + invert
    out = ds.filter(expr=~col("x").is_in([0, 1, 2, 3, 4, 5, 6, 7, 8, 9,
                                          10, 11, 12, 13, 14, 15, 16,
                                          17])).take_all()
    assert sorted(r["x"] for r in out) == [18, 19]
