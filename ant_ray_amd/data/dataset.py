"""Dataset: lazy logical plan over blocks + consumption APIs.

Role parity: reference python/ray/data/dataset.py (map_batches :467,
streaming_split :1881, iter_batches via iterator.py). Laziness: transforms
append logical ops; consumption (iter/take/write/materialize) runs the
streaming executor.
"""
from __future__ import annotations

import itertools
from typing import Any, Callable, Dict, Iterator, List, Optional, Union

import numpy as np
import pyarrow as pa

import ant_ray_amd as ray
from ant_ray_amd.data.block import Block, BlockAccessor, block_from_dict
from ant_ray_amd.data.context import DataContext
from ant_ray_amd.data.executor import execute_plan
from ant_ray_amd.data.plan import (
    ActorPoolStrategy,
    AllToAllOp,
    LimitOp,
    MapOp,
    ReadOp,
)


class Dataset:
    def __init__(self, ops: List[Any]):
        self._ops = ops

    # ------------------------------------------------------------ transforms

    def _with(self, op) -> "Dataset":
        return Dataset(self._ops + [op])

    def map_batches(
        self,
        fn: Union[Callable, type],
        *,
        batch_size: Optional[int] = "default",
        batch_format: Optional[str] = "default",
        compute: Optional[Any] = None,
        fn_args: tuple = (),
        fn_kwargs: Optional[dict] = None,
        fn_constructor_args: tuple = (),
        fn_constructor_kwargs: Optional[dict] = None,
        num_cpus: Optional[float] = None,
        num_gpus: Optional[float] = None,
        concurrency: Optional[Union[int, tuple]] = None,
        zero_copy_batch: bool = False,
        **_ignored,
    ) -> "Dataset":
        if batch_size == "default":
            batch_size = 1024
        import inspect

        if compute is None and concurrency is not None and inspect.isclass(fn):
            size = concurrency if isinstance(concurrency, int) else concurrency[-1]
            compute = ActorPoolStrategy(size=size)
        return self._with(MapOp(
            name=f"MapBatches({getattr(fn, '__name__', type(fn).__name__)})",
            kind="map_batches", fn=fn, batch_size=batch_size,
            batch_format=batch_format, fn_args=fn_args,
            fn_kwargs=fn_kwargs or {}, fn_constructor_args=fn_constructor_args,
            fn_constructor_kwargs=fn_constructor_kwargs or {},
            compute=compute, num_cpus=num_cpus, num_gpus=num_gpus,
            concurrency=concurrency,
        ))

    def map(self, fn, *, compute=None, num_cpus=None, num_gpus=None, **_):
        return self._with(MapOp(name="Map", kind="map_rows", fn=fn,
                                compute=compute, num_cpus=num_cpus,
                                num_gpus=num_gpus))

    def flat_map(self, fn, **kw):
        return self._with(MapOp(name="FlatMap", kind="flat_map", fn=fn))

    def filter(self, fn, **kw):
        return self._with(MapOp(name="Filter", kind="filter", fn=fn))

    def add_column(self, col: str, fn, **kw):
        def _add(batch):
            batch[col] = fn(batch)
            return batch

        return self.map_batches(_add, batch_format="pandas")

    def drop_columns(self, cols: List[str], **kw):
        def _drop(t: pa.Table):
            keep = [c for c in t.column_names if c not in cols]
            return t.select(keep)

        return self.map_batches(_drop, batch_format="pyarrow", batch_size=None)

    def select_columns(self, cols: List[str], **kw):
        return self._with(MapOp(
            name=f"Select{cols}", kind="map_batches",
            fn=lambda t: t.select(cols), batch_format="pyarrow",
            batch_size=None, meta={"type": "select", "columns": list(cols)}))

    def rename_columns(self, mapping: Dict[str, str], **kw):
        def _ren(t: pa.Table):
            return t.rename_columns([mapping.get(c, c) for c in t.column_names])

        return self.map_batches(_ren, batch_format="pyarrow", batch_size=None)

    def with_columns(self, exprs: Dict[str, Any]) -> "Dataset":
        """Add/replace columns from expressions (parity dataset.with_columns
        + data/expressions.py)."""
        from ant_ray_amd.data.expressions import Expr, eval_expr_to_column

        def _apply(t: pa.Table):
            for name, e in exprs.items():
                colv = eval_expr_to_column(t, e) if isinstance(e, Expr) else e
                if name in t.column_names:
                    t = t.set_column(t.column_names.index(name), name, colv)
                else:
                    t = t.append_column(name, colv)
            return t

        return self._with(MapOp(
            name="WithColumns", kind="map_batches", fn=_apply,
            batch_format="pyarrow", batch_size=None,
            meta={"type": "with_columns", "exprs": dict(exprs)}))

    def filter_expr(self, expr) -> "Dataset":
        """Vectorized filter by expression (the reference overloads
        Dataset.filter with Expr; kept as a separate method here)."""
        from ant_ray_amd.data.expressions import eval_expr_to_column

        def _apply(t: pa.Table):
            mask = eval_expr_to_column(t, expr)
            return t.filter(mask)

        return self._with(MapOp(
            name="FilterExpr", kind="map_batches", fn=_apply,
            batch_format="pyarrow", batch_size=None,
            meta={"type": "filter_expr", "expr": expr}))

    def limit(self, n: int) -> "Dataset":
        return self._with(LimitOp(name=f"Limit[{n}]", limit=n))

    def repartition(self, num_blocks: int, **kw) -> "Dataset":
        def _repart(refs: List[Any]) -> List[Any]:
            from ant_ray_amd.data.exchange import exchange

            P = max(1, num_blocks)

            def partitioner(block):
                n = block.num_rows
                per = -(-n // P) if n else 0
                return [block.slice(min(p * per, n), per) for p in range(P)]

            return exchange(refs, P, partitioner, lambda t: t)

        return self._with(AllToAllOp(name="Repartition", fn=_repart))

    def random_shuffle(self, *, seed: Optional[int] = None, **kw) -> "Dataset":
        def _shuffle(refs: List[Any]) -> List[Any]:
            from ant_ray_amd.data.exchange import shuffle_exchange

            return shuffle_exchange(refs, seed)

        return self._with(AllToAllOp(name="RandomShuffle", fn=_shuffle))

    def sort(self, key: Union[str, List[str]], descending: bool = False) -> "Dataset":
        keys = [key] if isinstance(key, str) else list(key)

        def _sort(refs: List[Any]) -> List[Any]:
            from ant_ray_amd.data.exchange import sort_exchange

            return sort_exchange(refs, keys, descending)

        return self._with(AllToAllOp(name="Sort", fn=_sort))

    def union(self, *others: "Dataset") -> "Dataset":
        me = self

        def _union(refs: List[Any]) -> List[Any]:
            out = list(refs)
            for o in others:
                out.extend(o.iter_internal_ref_bundles())
            return out

        return self._with(AllToAllOp(name="Union", fn=_union))

    def join(self, other: "Dataset", *, on: Union[str, List[str]],
             join_type: str = "inner",
             right_on: Optional[Union[str, List[str]]] = None,
             suffixes: tuple = ("", "_r")) -> "Dataset":
        """Hash join with another dataset on key column(s) (parity:
        reference data/_internal/execution/operators/join.py; arrow hash
        join inside; join_type: inner/left outer/right outer/full outer)."""
        keys = [on] if isinstance(on, str) else list(on)
        rkeys = ([right_on] if isinstance(right_on, str)
                 else list(right_on) if right_on else keys)
        jt = {"inner": "inner", "left": "left outer", "right": "right outer",
              "full": "full outer"}.get(join_type, join_type)

        def _join(refs: List[Any]) -> List[Any]:
            from ant_ray_amd.data.exchange import join_exchange

            right_refs = list(other.iter_internal_ref_bundles())
            if not refs or not right_refs:
                if jt == "inner":
                    return []
                # outer joins with one empty side degrade to the other side
                return list(refs) or right_refs
            return join_exchange(refs, right_refs, keys, rkeys, jt, suffixes)

        return self._with(AllToAllOp(name=f"Join[{jt}]", fn=_join))

    def zip(self, other: "Dataset") -> "Dataset":
        """Column-wise zip of two same-length datasets (parity:
        Dataset.zip — duplicate column names from the right side get a
        _1 suffix)."""

        def _zip(refs: List[Any]) -> List[Any]:
            lt = pa.concat_tables([ray.get(r) for r in refs],
                                  promote_options="default")
            rblocks = [ray.get(r) for r in other.iter_internal_ref_bundles()]
            rt = pa.concat_tables(rblocks, promote_options="default")
            if lt.num_rows != rt.num_rows:
                raise ValueError(
                    f"zip needs equal row counts: {lt.num_rows} vs {rt.num_rows}")
            cols = {name: lt.column(name) for name in lt.column_names}
            for name in rt.column_names:
                out = name if name not in cols else f"{name}_1"
                cols[out] = rt.column(name)
            res = pa.table(cols)
            k = max(len(refs), 1)
            per = max(1, -(-res.num_rows // k))
            return [ray.put(res.slice(s, per))
                    for s in range(0, res.num_rows, per)]

        return self._with(AllToAllOp(name="Zip", fn=_zip))

    def unique(self, column: str) -> List[Any]:
        """Distinct values of a column (parity: Dataset.unique)."""
        vals = set()
        for block in self.iter_blocks():
            vals.update(block.column(column).to_pylist())
        return sorted(vals, key=lambda v: (v is None, v))

    def groupby(self, key: str) -> "GroupedData":
        return GroupedData(self, key)

    def random_sample(self, fraction: float, *, seed=None) -> "Dataset":
        rng = np.random.RandomState(seed)

        def _sample(t: pa.Table):
            mask = rng.rand(t.num_rows) < fraction
            return t.filter(pa.array(mask))

        return self.map_batches(_sample, batch_format="pyarrow", batch_size=None)

    # ----------------------------------------------------------- consumption

    def iter_internal_ref_bundles(self) -> Iterator[Any]:
        return execute_plan(self._ops)

    def iter_blocks(self) -> Iterator[Block]:
        for ref in self.iter_internal_ref_bundles():
            yield ray.get(ref)

    def take(self, limit: int = 20) -> List[Dict[str, Any]]:
        out = []
        for block in self.limit(limit).iter_blocks():
            out.extend(BlockAccessor(block).iter_rows())
            if len(out) >= limit:
                break
        return out[:limit]

    def take_all(self) -> List[Dict[str, Any]]:
        out = []
        for block in self.iter_blocks():
            out.extend(BlockAccessor(block).iter_rows())
        return out

    def take_batch(self, batch_size: int = 20, *, batch_format="default"):
        for batch in self.iter_batches(batch_size=batch_size,
                                       batch_format=batch_format):
            return batch
        return {}

    def show(self, limit: int = 20):
        for row in self.take(limit):
            print(row)

    def count(self) -> int:
        ops = self._ops
        if len(ops) == 1 and isinstance(ops[0], ReadOp) and ops[0].num_rows is not None:
            return ops[0].num_rows

        @ray.remote(num_cpus=0.25)
        def _rows(b):
            return BlockAccessor(b).num_rows()

        # counts happen task-side: only ints travel to the driver
        return sum(ray.get([_rows.remote(r)
                            for r in self.iter_internal_ref_bundles()]))

    def schema(self):
        for block in self.iter_blocks():
            return Schema(BlockAccessor(block).schema())
        return None

    def columns(self) -> List[str]:
        s = self.schema()
        return list(s.names) if s is not None else []

    def iter_rows(self) -> Iterator[Dict[str, Any]]:
        for block in self.iter_blocks():
            yield from BlockAccessor(block).iter_rows()

    def iter_batches(self, *, batch_size: Optional[int] = 256,
                     batch_format: str = "default", drop_last: bool = False,
                     prefetch_batches: int = 1, **_) -> Iterator[Any]:
        from ant_ray_amd.data.iterator import rebatch_blocks

        yield from rebatch_blocks(self.iter_blocks(), batch_size, batch_format,
                                  drop_last)

    def iter_torch_batches(self, *, batch_size: Optional[int] = 256,
                           dtypes=None, device: Optional[str] = None,
                           collate_fn: Optional[Callable] = None,
                           drop_last: bool = False, **kw) -> Iterator[Any]:
        from ant_ray_amd.data.iterator import to_torch_batch

        for batch in self.iter_batches(batch_size=batch_size, drop_last=drop_last):
            yield to_torch_batch(batch, dtypes, device, collate_fn)

    # ------------------------------------------------- aggregates / stats
    def _column_array(self, on: str):
        import pyarrow as pa

        blocks = list(self.iter_blocks())
        if not blocks:
            return None
        t = pa.concat_tables(blocks, promote_options="default")
        return t.column(on).combine_chunks()

    def aggregate(self, *aggs):
        """Run AggregateFn-style aggregations (parity Dataset.aggregate);
        also accepts ("name", col) pairs for sum/min/max/mean/std."""
        out = {}
        for a in aggs:
            if isinstance(a, tuple) and len(a) == 2:
                name, col = a
                out[f"{name}({col})"] = getattr(self, name)(col)
            else:  # AggregateFn-like: init/accumulate_block/merge/name
                acc = a.init(None)
                for b in self.iter_blocks():
                    acc = a.merge(acc, a.accumulate_block(a.init(None), b))
                out[getattr(a, "name", repr(a))] = (
                    a.finalize(acc) if hasattr(a, "finalize") else acc)
        return out

    def sum(self, on: str):
        import pyarrow.compute as pc

        arr = self._column_array(on)
        return None if arr is None else pc.sum(arr).as_py()

    def min(self, on: str):
        import pyarrow.compute as pc

        arr = self._column_array(on)
        return None if arr is None else pc.min(arr).as_py()

    def max(self, on: str):
        import pyarrow.compute as pc

        arr = self._column_array(on)
        return None if arr is None else pc.max(arr).as_py()

    def mean(self, on: str):
        import pyarrow.compute as pc

        arr = self._column_array(on)
        return None if arr is None else pc.mean(arr).as_py()

    def std(self, on: str, ddof: int = 1):
        import pyarrow.compute as pc

        arr = self._column_array(on)
        return None if arr is None else pc.stddev(arr, ddof=ddof).as_py()

    # ---------------------------------------------------- splits / order
    def split_at_indices(self, indices: List[int]) -> List["Dataset"]:
        """Row-exact split (parity Dataset.split_at_indices)."""
        import pyarrow as pa

        blocks = list(self.iter_blocks())
        table = (pa.concat_tables(blocks, promote_options="default")
                 if blocks else pa.table({}))
        bounds = [0] + list(indices) + [table.num_rows]
        out = []
        import ant_ray_amd as ray

        for s0, e0 in zip(bounds[:-1], bounds[1:]):
            piece = table.slice(s0, max(0, e0 - s0))
            out.append(MaterializedDataset([ray.put(piece)]))
        return out

    def split_proportionately(self, proportions: List[float]) -> List["Dataset"]:
        n = self.count()
        idx, acc = [], 0.0
        for p in proportions:
            acc += p
            idx.append(int(n * acc))
        return self.split_at_indices(idx)

    def train_test_split(self, test_size: float, *, shuffle: bool = False,
                         seed=None) -> List["Dataset"]:
        ds = self.random_shuffle(seed=seed) if shuffle else self
        return ds.split_proportionately([1.0 - test_size])

    streaming_train_test_split = train_test_split

    def randomize_block_order(self, *, seed: Optional[int] = None) -> "Dataset":
        import random as _random

        def _shuf(refs: List[Any]) -> List[Any]:
            refs = list(refs)
            _random.Random(seed).shuffle(refs)
            return refs

        return self._with(AllToAllOp(name="RandomizeBlockOrder", fn=_shuf))

    def with_column(self, name: str, expr) -> "Dataset":
        return self.with_columns({name: expr})

    # -------------------------------------------------- metadata / plumbing
    def name(self) -> Optional[str]:
        return getattr(self, "_name", None)

    def set_name(self, name: Optional[str]):
        self._name = name

    def get_dataset_id(self) -> str:
        if not hasattr(self, "_dataset_id"):
            import uuid

            self._dataset_id = uuid.uuid4().hex
        return self._dataset_id

    def context(self):
        from ant_ray_amd.data.context import DataContext

        return DataContext.get_current()

    def copy(self) -> "Dataset":
        return Dataset(list(self._ops))

    def explain(self) -> str:
        """Logical plan rendering (parity Dataset.explain)."""
        lines = ["Execution plan:"]
        for op in self._ops:
            lines.append(f"  {type(op).__name__}: {getattr(op, 'name', '')}")
        return "\n".join(lines)

    def names(self) -> List[str]:
        return self.columns()

    def types(self) -> List[Any]:
        sch = self.schema()
        return list(getattr(sch, "types", []))

    def input_files(self) -> List[str]:
        op = self._ops[0] if self._ops else None
        return list(getattr(op, "input_files", []) or [])

    def get_internal_block_refs(self) -> List[Any]:
        return list(self.iter_internal_ref_bundles())

    def iterator(self):
        """DataIterator over this dataset (parity Dataset.iterator)."""
        return self.split(1)[0]

    def to_numpy_refs(self) -> List[Any]:
        import ant_ray_amd as ray

        out = []
        for b in self.iter_blocks():
            out.append(ray.put(
                {c: b.column(c).combine_chunks().to_numpy(zero_copy_only=False)
                 for c in b.column_names}))
        return out

    def to_pandas_refs(self) -> List[Any]:
        import ant_ray_amd as ray

        return [ray.put(b.to_pandas()) for b in self.iter_blocks()]

    def write_numpy(self, path: str, *, column: str = "data", **_):
        """Write each block's `column` as an .npy file under path."""
        import os

        import numpy as np

        os.makedirs(path, exist_ok=True)
        for i, b in enumerate(self.iter_blocks()):
            arr = b.column(column).combine_chunks().to_numpy(
                zero_copy_only=False)
            np.save(os.path.join(path, f"block_{i:05d}.npy"), arr)

    def write_sql(self, sql: str, connection_factory, **_):
        """INSERT each row via DB-API (parity Dataset.write_sql; pairs
        with read_sql). `sql` must be an INSERT with placeholders."""
        conn = connection_factory()
        cur = conn.cursor()
        for b in self.iter_blocks():
            cols = b.column_names
            for row in b.to_pylist():
                cur.execute(sql, tuple(row[c] for c in cols))
        conn.commit()

    def _no_lib(self, lib: str):
        raise ImportError(
            f"Dataset conversion requires {lib}, which is not installed in "
            f"this air-gapped image (reference parity: the method exists "
            f"and delegates to {lib}).")

    def to_tf(self, *a, **k):
        self._no_lib("tensorflow")

    def iter_tf_batches(self, *a, **k):
        self._no_lib("tensorflow")

    def to_dask(self, *a, **k):
        self._no_lib("dask")

    def to_modin(self, *a, **k):
        self._no_lib("modin")

    def to_mars(self, *a, **k):
        self._no_lib("mars")

    def to_spark(self, *a, **k):
        self._no_lib("pyspark")

    def to_daft(self, *a, **k):
        self._no_lib("daft")

    def to_random_access_dataset(self, *a, **k):
        self._no_lib("random-access datasets (reference experimental)")

    def to_pandas(self, limit: Optional[int] = None):
        import pandas as pd

        blocks = list((self.limit(limit) if limit else self).iter_blocks())
        if not blocks:
            return pd.DataFrame()
        return pa.concat_tables(blocks, promote_options="default").to_pandas()

    def to_arrow_refs(self) -> List[Any]:
        return list(self.iter_internal_ref_bundles())

    def materialize(self) -> "MaterializedDataset":
        refs = list(self.iter_internal_ref_bundles())
        return MaterializedDataset(refs)

    def stats(self) -> str:
        return f"Dataset(ops={[getattr(o, 'name', o) for o in self._ops]})"

    def num_blocks(self) -> int:
        return len(list(self.iter_internal_ref_bundles()))

    def size_bytes(self) -> int:
        @ray.remote(num_cpus=0.25)
        def _sz(b):
            return BlockAccessor(b).size_bytes()

        return sum(ray.get([_sz.remote(r)
                            for r in self.iter_internal_ref_bundles()]))

    # ----------------------------------------------------------------- split

    def split(self, n: int, *, equal: bool = False, locality_hints=None
              ) -> List["MaterializedDataset"]:
        refs = list(self.iter_internal_ref_bundles())
        if equal:
            blocks = [ray.get(r) for r in refs]
            table = (pa.concat_tables(blocks, promote_options="default")
                     if blocks else pa.table({}))
            per = table.num_rows // n  # equal shards; remainder rows dropped
            return [
                MaterializedDataset([ray.put(table.slice(i * per, per))])
                for i in range(n)
            ]
        shards: List[List[Any]] = [[] for _ in range(n)]
        for i, r in enumerate(refs):
            shards[i % n].append(r)
        return [MaterializedDataset(s) for s in shards]

    def train_test_split(self, test_size: float, *, shuffle: bool = False,
                         seed=None):
        ds = self.random_shuffle(seed=seed) if shuffle else self
        rows = ds.take_all()
        n_test = int(len(rows) * test_size)
        from ant_ray_amd.data import from_items

        return from_items(rows[: len(rows) - n_test]), from_items(
            rows[len(rows) - n_test:])

    def streaming_split(self, n: int, *, equal: bool = False,
                        locality_hints=None) -> List[Any]:
        """N coordinated iterators over one execution (parity dataset.py:1881;
        used by Train's get_dataset_shard)."""
        from ant_ray_amd.data.iterator import SplitCoordinator, DataIterator

        coord = ray.remote(max_concurrency=2 * n + 2)(SplitCoordinator).remote(
            self, n, equal
        )
        return [DataIterator(coord, i) for i in range(n)]

    # ----------------------------------------------------------------- write

    def write_parquet(self, path: str, **kw):
        self._write(path, "parquet")

    def write_csv(self, path: str, **kw):
        self._write(path, "csv")

    def write_json(self, path: str, **kw):
        self._write(path, "json")

    def _write(self, path: str, fmt: str):
        import os

        os.makedirs(path, exist_ok=True)
        refs = []
        write_remote = ray.remote(_write_block)
        for i, ref in enumerate(self.iter_internal_ref_bundles()):
            refs.append(write_remote.remote(ref, path, fmt, i))
        ray.get(refs)

    def write_datasink(self, datasink, *, concurrency=None, **_):
        """Write through a custom Datasink (parity: reference
        Dataset.write_datasink → datasink.py lifecycle): driver runs
        on_write_start, one Ray task per block bundle runs write(),
        driver finishes with on_write_complete / on_write_failed."""
        datasink.on_write_start()

        def _task(block, sink, idx):
            return sink.write([block], {"task_index": idx})

        try:
            w = ray.remote(_task)
            results = ray.get([w.remote(ref, datasink, i)
                               for i, ref in
                               enumerate(self.iter_internal_ref_bundles())])
        except Exception as e:
            datasink.on_write_failed(e)
            raise
        datasink.on_write_complete(results)
        return results

    def summary(self, *, columns=None):
        """Per-column statistics (parity: reference Dataset.summary →
        data/stats.py DatasetSummary): count/min/max/mean/std for numeric
        columns, count/missing for the rest."""
        from ant_ray_amd.data.stats import DatasetSummary, _summarize_blocks

        schema = self.schema()
        cols = columns or (schema.names if schema else [])
        stats = _summarize_blocks(self, cols)
        return DatasetSummary(dataset_schema=schema, columns=cols,
                              stats=stats)

    def __repr__(self):
        return self.stats()


def _write_block(block, path, fmt, index):
    import os

    import pyarrow.csv as pcsv
    import pyarrow.parquet as pq

    f = os.path.join(path, f"part-{index:06d}.{fmt}")
    if fmt == "parquet":
        pq.write_table(block, f)
    elif fmt == "csv":
        pcsv.write_csv(block, f)
    elif fmt == "json":
        import json

        rows = list(BlockAccessor(block).iter_rows())
        with open(f, "w") as fh:
            for r in rows:
                fh.write(json.dumps({k: _json_safe(v) for k, v in r.items()}) + "\n")
    return f


def _json_safe(v):
    if isinstance(v, (np.integer,)):
        return int(v)
    if isinstance(v, (np.floating,)):
        return float(v)
    if isinstance(v, np.ndarray):
        return v.tolist()
    return v


class MaterializedDataset(Dataset):
    """A Dataset whose blocks are already in the object store."""

    def __init__(self, refs: List[Any]):
        super().__init__([
            ReadOp(name="Materialized", read_tasks=[], num_rows=None),
            AllToAllOp(name="Materialized", fn=lambda _refs, refs=refs: refs),
        ])
        self._refs = refs


class GroupedData:
    """Minimal groupby: aggregates + map_groups (parity data/grouped_data.py)."""

    def __init__(self, ds: Dataset, key: str):
        self.ds = ds
        self.key = key

    def _agg(self, aggs: List[tuple]) -> Dataset:
        key = self.key

        def _do(refs: List[Any]) -> List[Any]:
            from ant_ray_amd.data.exchange import groupby_exchange

            def finalize(t: pa.Table):
                if t.num_rows == 0:
                    return t
                # arrow names output "<col>_<agg>"; keep as-is
                return t.group_by(key).aggregate(aggs)

            # hash exchange: a group lands wholly in one reducer, so the
            # per-partition aggregates concatenate into the global answer
            return groupby_exchange(refs, key, finalize)

        return self.ds._with(AllToAllOp(name="GroupByAgg", fn=_do))

    def count(self) -> Dataset:
        return self._agg([(self.key, "count")])

    def sum(self, col: str) -> Dataset:
        return self._agg([(col, "sum")])

    def mean(self, col: str) -> Dataset:
        return self._agg([(col, "mean")])

    def min(self, col: str) -> Dataset:
        return self._agg([(col, "min")])

    def max(self, col: str) -> Dataset:
        return self._agg([(col, "max")])

    def map_groups(self, fn, *, batch_format="pandas") -> Dataset:
        key = self.key

        def _do(refs: List[Any]) -> List[Any]:
            from ant_ray_amd.data.exchange import groupby_exchange

            def finalize(t: pa.Table):
                if t.num_rows == 0:
                    return t
                df = t.to_pandas()
                outs = []
                for _, grp in df.groupby(key):
                    res = fn(grp if batch_format == "pandas" else
                             {c: grp[c].to_numpy() for c in grp.columns})
                    outs.append(BlockAccessor.for_block(res).to_arrow())
                return (pa.concat_tables(outs, promote_options="default")
                        if outs else t.slice(0, 0))

            return groupby_exchange(refs, key, finalize)

        return self.ds._with(AllToAllOp(name="MapGroups", fn=_do))


class Schema:
    """Dataset schema wrapper (parity: reference data/dataset.py Schema —
    .names / .types over the underlying pyarrow schema)."""

    def __init__(self, base_schema):
        self.base_schema = base_schema

    @property
    def names(self):
        return list(self.base_schema.names)

    @property
    def types(self):
        return list(self.base_schema.types)

    def __eq__(self, other):
        o = other.base_schema if isinstance(other, Schema) else other
        return self.base_schema == o

    def __repr__(self):
        cols = ", ".join(f"{n}: {t}" for n, t in
                         zip(self.names, self.types))
        return f"Schema({cols})"
