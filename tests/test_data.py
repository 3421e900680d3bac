"""Ray Data parity tests on CPU: creation, transforms, execution, splits,
writes, and Train integration (streaming_split shards)."""
import os

import numpy as np
import pytest


@pytest.fixture(scope="module")
def ray_mod():
    import ant_ray_amd as ray

    if ray.is_initialized():
        ray.shutdown()  # never inherit another module's (possibly dying) session
    if not ray.is_initialized():
        ray.init(num_cpus=8)
    yield ray
    ray.shutdown()


@pytest.fixture(scope="module")
def data(ray_mod):
    import ant_ray_amd.data as data

    return data


def test_range_count_take(data):
    ds = data.range(100)
    assert ds.count() == 100
    rows = ds.take(5)
    assert rows == [{"id": i} for i in range(5)]


def test_map_batches_numpy(data):
    ds = data.range(1000).map_batches(lambda b: {"id": b["id"] * 2})
    out = ds.take_all()
    assert len(out) == 1000
    assert sorted(r["id"] for r in out) == [2 * i for i in range(1000)]


def test_map_filter_flatmap_fusion(data):
    ds = (data.range(50)
          .map(lambda r: {"id": r["id"] + 1})
          .filter(lambda r: r["id"] % 2 == 0)
          .flat_map(lambda r: [{"v": r["id"]}, {"v": -r["id"]}]))
    vals = sorted(r["v"] for r in ds.take_all())
    expect = sorted(v for i in range(50) if (i + 1) % 2 == 0 for v in ((i + 1), -(i + 1)))
    assert vals == expect


def test_map_batches_actor_class(data):
    class AddConst:
        def __init__(self, c):
            self.c = c

        def __call__(self, batch):
            return {"id": batch["id"] + self.c}

    ds = data.range(100).map_batches(
        AddConst, fn_constructor_args=(7,), concurrency=2, batch_size=32,
    )
    out = sorted(r["id"] for r in ds.take_all())
    assert out == [i + 7 for i in range(100)]


def test_batch_formats(data):
    import pandas as pd
    import pyarrow as pa

    ds = data.range(10)

    def check_pandas(df):
        assert isinstance(df, pd.DataFrame), type(df)
        df["is_df"] = 1
        return df

    def check_arrow(t):
        assert isinstance(t, pa.Table), type(t)
        return t.append_column("is_tbl", pa.array([1] * t.num_rows))

    assert all(r["is_df"] == 1 for r in
               ds.map_batches(check_pandas, batch_format="pandas").take_all())
    assert all(r["is_tbl"] == 1 for r in
               ds.map_batches(check_arrow, batch_format="pyarrow").take_all())


def test_sort_shuffle_repartition(data):
    ds = data.range(200).random_shuffle(seed=4)
    ids = [r["id"] for r in ds.take_all()]
    assert ids != list(range(200)) and sorted(ids) == list(range(200))
    ds2 = data.from_items([{"k": i % 5, "v": i} for i in range(100)]).sort(
        "v", descending=True)
    vs = [r["v"] for r in ds2.take_all()]
    assert vs == sorted(vs, reverse=True)
    assert data.range(100).repartition(7).num_blocks() == 7


def test_groupby(data):
    ds = data.from_items([{"k": i % 3, "v": float(i)} for i in range(30)])
    res = {r["k"]: r["v_sum"] for r in ds.groupby("k").sum("v").take_all()}
    expect = {}
    for i in range(30):
        expect[i % 3] = expect.get(i % 3, 0.0) + i
    assert res == expect


def test_iter_batches_exact_sizes(data):
    ds = data.range(100, override_num_blocks=7)
    sizes = [len(b["id"]) for b in ds.iter_batches(batch_size=32)]
    assert sizes == [32, 32, 32, 4]
    sizes = [len(b["id"]) for b in ds.iter_batches(batch_size=32, drop_last=True)]
    assert sizes == [32, 32, 32]


def test_iter_torch_batches(data):
    import torch

    ds = data.range(16)
    batches = list(ds.iter_torch_batches(batch_size=8, dtypes=torch.float32))
    assert len(batches) == 2
    assert batches[0]["id"].dtype == torch.float32
    assert batches[0]["id"].shape == (8,)


def test_tensor_columns(data):
    arr = np.random.rand(32, 4, 4).astype(np.float32)
    ds = data.from_numpy(arr)
    got = ds.take_batch(32)["data"]
    np.testing.assert_allclose(got.reshape(32, 4, 4), arr)


def test_write_read_roundtrip(data, tmp_path):
    p = str(tmp_path / "pq")
    data.range(100).map_batches(lambda b: {"id": b["id"], "sq": b["id"] ** 2}).write_parquet(p)
    back = data.read_parquet(p)
    rows = back.take_all()
    assert len(rows) == 100
    assert all(r["sq"] == r["id"] ** 2 for r in rows)
    c = str(tmp_path / "csv")
    data.range(10).write_csv(c)
    assert data.read_csv(c).count() == 10


def test_split_and_streaming_split(data):
    ds = data.range(100, override_num_blocks=10)
    shards = ds.split(4)
    assert sum(s.count() for s in shards) == 100
    its = ds.streaming_split(2, equal=True)
    seen = [sorted(r["id"] for r in it.iter_rows()) for it in its]
    assert len(seen[0]) + len(seen[1]) == 100
    assert not (set(seen[0]) & set(seen[1]))


def test_limit_streams_early(data):
    ds = data.range(10_000, override_num_blocks=100)
    assert len(ds.take(30)) == 30


def test_train_get_dataset_shard(data, ray_mod):
    from ant_ray_amd.train import ScalingConfig
    from ant_ray_amd.train.torch import TorchTrainer

    ds = data.range(64, override_num_blocks=8)

    def train_fn(config):
        import torch
        import torch.distributed as dist

        from ant_ray_amd import train

        shard = train.get_dataset_shard("train")
        total = 0
        for batch in shard.iter_torch_batches(batch_size=8):
            total += len(batch["id"])
        t = torch.tensor([total])
        dist.all_reduce(t)
        train.report({"rows": total, "total_rows": int(t.item())})

    trainer = TorchTrainer(
        train_fn,
        scaling_config=ScalingConfig(num_workers=2),
        datasets={"train": ds},
    )
    result = trainer.fit()
    # every row lands on exactly one worker across the split
    assert result.metrics["total_rows"] == 64
    assert result.metrics["rows"] == 32  # 8 equal blocks dealt round-robin


def test_expressions(data):
    from ant_ray_amd.data import col, lit

    ds = data.range(20).with_columns({
        "double": col("id") * 2,
        "shifted": col("id") + lit(100),
    })
    rows = ds.take_all()
    assert all(r["double"] == 2 * r["id"] and r["shifted"] == r["id"] + 100
               for r in rows)
    kept = ds.filter_expr((col("id") > 5) & (col("id") <= 10)).take_all()
    assert sorted(r["id"] for r in kept) == [6, 7, 8, 9, 10]


def test_join_and_zip(data):
    left = data.from_items([{"id": i, "a": i * 10} for i in range(6)])
    right = data.from_items([{"id": i, "b": i * 100} for i in range(3, 9)])

    joined = left.join(right, on="id").sort("id").take_all()
    assert [r["id"] for r in joined] == [3, 4, 5]
    assert joined[0]["a"] == 30 and joined[0]["b"] == 300

    outer = left.join(right, on="id", join_type="left").count()
    assert outer == 6

    z = left.zip(data.from_items([{"c": i} for i in range(6)])).take_all()
    assert z[0]["a"] == 0 and "c" in z[0]
    assert len(z) == 6

    assert left.unique("id") == [0, 1, 2, 3, 4, 5]


def test_custom_datasource_and_datasink(ray_mod, tmp_path):
    """Datasource/ReadTask → read_datasource; RowBased/BlockBased file
    datasinks with SaveMode semantics (parity: data/datasource/)."""
    import os

    import ant_ray_amd.data as rd
    from ant_ray_amd.data import (Datasource, ReadTask,
                                  RowBasedFileDatasink, SaveMode)

    class SquaresSource(Datasource):
        def __init__(self, n):
            self.n = n

        def get_read_tasks(self, parallelism):
            per = max(1, self.n // parallelism)
            tasks = []
            for s in range(0, self.n, per):
                e = min(s + per, self.n)
                tasks.append(ReadTask(
                    lambda s=s, e=e: [{"id": list(range(s, e)),
                                       "sq": [i * i for i in range(s, e)]}],
                    metadata={"num_rows": e - s}))
            return tasks

    ds = rd.read_datasource(SquaresSource(100), parallelism=5)
    assert ds.count() == 100
    assert sorted(r["sq"] for r in ds.take_all())[-1] == 99 * 99

    class CsvRowSink(RowBasedFileDatasink):
        def write_row_to_file(self, row, file):
            file.write(f"{row['id']},{row['sq']}\n".encode())

    out = str(tmp_path / "sink_out")
    sink = CsvRowSink(out, file_format="csv")
    ds.write_datasink(sink)
    lines = []
    for f in os.listdir(out):
        with open(os.path.join(out, f)) as fh:
            lines += [ln for ln in fh.read().splitlines() if ln]
    assert len(lines) == 100

    # SaveMode.ERROR refuses a non-empty dir
    with pytest.raises(ValueError):
        rd.range(3).write_datasink(CsvRowSink(out, file_format="csv",
                                              mode=SaveMode.ERROR))


def test_from_refs_read_sql_summary(ray_mod, tmp_path):
    """from_pandas_refs/from_numpy_refs/from_arrow_refs, read_sql over
    sqlite3, Dataset.summary, Preprocessor (parity: data/read_api.py,
    stats.py, preprocessor.py)."""
    import sqlite3

    import numpy as np
    import pandas as pd
    import pyarrow as pa

    import ant_ray_amd as ray
    import ant_ray_amd.data as rd

    dfs = [pd.DataFrame({"x": [1, 2]}), pd.DataFrame({"x": [3, 4]})]
    ds = rd.from_pandas_refs([ray.put(d) for d in dfs])
    assert sorted(r["x"] for r in ds.take_all()) == [1, 2, 3, 4]

    ds = rd.from_numpy_refs([ray.put(np.arange(3))], column="v")
    assert sorted(r["v"] for r in ds.take_all()) == [0, 1, 2]

    ds = rd.from_arrow_refs([ray.put(pa.table({"y": [7, 8]}))])
    assert sorted(r["y"] for r in ds.take_all()) == [7, 8]

    # read_sql against sqlite
    db = str(tmp_path / "t.db")
    conn = sqlite3.connect(db)
    conn.execute("create table t (a int, b real)")
    conn.executemany("insert into t values (?, ?)",
                     [(i, i * 0.5) for i in range(10)])
    conn.commit()
    conn.close()
    ds = rd.read_sql("select * from t where a >= 2",
                     lambda: sqlite3.connect(db))
    assert ds.count() == 8 and ds.schema().names == ["a", "b"]

    # summary stats
    summ = rd.range(100).summary()
    st = summ.column_stats("id")
    assert st["count"] == 100 and st["min"] == 0 and st["max"] == 99
    assert abs(st["mean"] - 49.5) < 1e-9
    assert "id" in summ.to_pandas().columns

    # Preprocessor
    from ant_ray_amd.data import Preprocessor

    class Center(Preprocessor):
        def _fit(self, ds):
            self.mean_ = ds.summary(columns=["id"]).column_stats("id")["mean"]

        def _transform_pandas(self, df):
            df["id"] = df["id"] - self.mean_
            return df

    pre = Center()
    with pytest.raises(Exception):
        pre.transform(rd.range(10))  # not fitted
    out = pre.fit_transform(rd.range(10))
    vals = sorted(r["id"] for r in out.take_all())
    assert abs(vals[0] + 4.5) < 1e-9 and abs(vals[-1] - 4.5) < 1e-9

    # unavailable readers raise informative errors
    with pytest.raises(NotImplementedError):
        rd.read_images("/tmp/x")


def test_expression_namespaces(ray_mod):
    import pyarrow as pa

    from ant_ray_amd import data
    from ant_ray_amd.data.expressions import col, lit

    ds = data.from_items([
        {"s": " Hello ", "n": -3.7, "tags": [1, 2, 3]},
        {"s": "world", "n": 2.2, "tags": [4]},
    ])
    out = ds.with_columns({
        "lower": col("s").str.strip().str.lower(),
        "has_l": col("s").str.contains("l"),
        "absn": col("n").abs().round(0),
        "ntags": col("tags").list.len(),
        "inset": col("n").round(0).cast(pa.int64()).is_in([-4, 2]),
    }).take_all()
    assert [r["lower"] for r in out] == ["hello", "world"]
    assert [r["absn"] for r in out] == [4.0, 2.0]
    assert [r["ntags"] for r in out] == [3, 1]
    assert [r["inset"] for r in out] == [True, True]
    # between + fill_null
    ds2 = data.from_items([{"x": 1}, {"x": None}, {"x": 9}])
    vals = ds2.with_columns({"ok": col("x").fill_null(0).between(0, 5)}
                            ).take_all()
    assert [r["ok"] for r in vals] == [True, True, False]


def test_parquet_projection_and_filter_pushdown(ray_mod, tmp_path):
    import pyarrow as pa
    import pyarrow.parquet as pq

    from ant_ray_amd import data
    from ant_ray_amd.data.expressions import col
    from ant_ray_amd.data.optimizer import optimize
    from ant_ray_amd.data.plan import ReadOp

    t = pa.table({"a": list(range(100)), "b": [f"s{i}" for i in range(100)],
                  "c": [i * 1.5 for i in range(100)]})
    pq.write_table(t, str(tmp_path / "f.parquet"))

    ds = (data.read_parquet(str(tmp_path / "f.parquet"))
          .filter_expr(col("a") < 10)
          .select_columns(["a", "c"]))
    # the optimizer folds both into the ReadOp
    opt = optimize(ds._ops)
    assert len(opt) == 1 and isinstance(opt[0], ReadOp)
    rows = ds.take_all()
    assert len(rows) == 10
    assert set(rows[0].keys()) == {"a", "c"}


def test_filter_reorder_before_with_columns(ray_mod):
    from ant_ray_amd import data
    from ant_ray_amd.data.expressions import col
    from ant_ray_amd.data.optimizer import filter_reorder

    ds = (data.from_items([{"x": i} for i in range(20)])
          .with_columns({"y": col("x") * 2})
          .filter_expr(col("x") > 15))
    ops = filter_reorder(ds._ops)
    metas = [(o.meta or {}).get("type") for o in ops[1:]]
    assert metas == ["filter_expr", "with_columns"]  # swapped
    # and NOT swapped when the filter uses the created column
    ds2 = (data.from_items([{"x": i} for i in range(20)])
           .with_columns({"y": col("x") * 2})
           .filter_expr(col("y") > 15))
    metas2 = [(o.meta or {}).get("type") for o in filter_reorder(ds2._ops)[1:]]
    assert metas2 == ["with_columns", "filter_expr"]
    assert len(ds.take_all()) == 4


def test_backpressure_policies():
    from ant_ray_amd.data.backpressure import (
        ConcurrencyCapBackpressurePolicy,
        ObjectStoreMemoryBackpressurePolicy,
    )

    cap = ConcurrencyCapBackpressurePolicy(cap=2)
    assert cap.can_add_input("s", 0) and cap.can_add_input("s", 1)
    assert not cap.can_add_input("s", 2)
    mem = ObjectStoreMemoryBackpressurePolicy(high_watermark=0.8)
    mem._last_frac = 0.9
    mem._last_poll = __import__("time").monotonic() + 100  # pin the poll
    assert mem.can_add_input("s", 0)  # always drainable
    assert not mem.can_add_input("s", 3)


def test_dataset_aggregates_and_splits(ray_mod):
    import ant_ray_amd.data as data

    ds = data.from_items([{"x": i, "y": float(i) * 2} for i in range(100)])
    assert ds.sum("x") == 4950
    assert abs(ds.mean("y") - 99.0) < 1e-6
    assert ds.max("x") == 99 and ds.min("x") == 0
    assert ds.std("x") > 0
    parts = ds.split_at_indices([30, 60])
    assert [p.count() for p in parts] == [30, 30, 40]
    tr, te = ds.train_test_split(0.2)
    assert (tr.count(), te.count()) == (80, 20)
    assert ds.randomize_block_order(seed=1).count() == 100
    assert ds.aggregate(("sum", "x"))["sum(x)"] == 4950
    assert "Execution plan" in ds.explain()
    assert ds.names() == ds.columns()
    ds.set_name("bench")
    assert ds.name() == "bench"
