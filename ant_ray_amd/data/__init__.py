"""ant_ray_amd.data — Ray Data parity: lazy, streaming datasets on blocks.

Role parity: reference python/ray/data/ (~145k LoC).
Creation APIs here cover the file formats available offline (parquet, csv,
json, text, binary, numpy, range, items, pandas, arrow, torch, huggingface);
cloud/warehouse sources (bigquery, mongo, …) are out of scope in this
air-gapped build and raise ImportError-style errors naming the source.
"""
from __future__ import annotations

import glob as _glob
import os
from dataclasses import dataclass as _dataclass, field as _field
from typing import Any, Dict, List, Optional

import numpy as np

from ant_ray_amd.data.block import Block, BlockAccessor, block_from_dict
from ant_ray_amd.data.context import DataContext
from ant_ray_amd.data.dataset import Dataset, MaterializedDataset
from ant_ray_amd.data.iterator import DataIterator
from ant_ray_amd.data.expressions import col, lit
from ant_ray_amd.data.plan import ActorPoolStrategy, ReadOp

from ant_ray_amd.data.datasource import (  # noqa: E402
    BlockBasedFileDatasink,
    Datasink,
    Datasource,
    ReadTask,
    RowBasedFileDatasink,
    SaveMode,
    SinkMode,
)
from ant_ray_amd.data.dataset import Schema  # noqa: E402
from ant_ray_amd.data.preprocessor import Preprocessor  # noqa: E402
from ant_ray_amd.data.stats import DatasetSummary  # noqa: E402

# reference-name aliases
DatasetContext = DataContext
DatasetIterator = DataIterator
NodeIdStr = str


@_dataclass
class ExecutionResources:
    """Resource budget for a Data execution (parity: reference
    execution_options.py ExecutionResources)."""

    cpu: Optional[float] = None
    gpu: Optional[float] = None
    object_store_memory: Optional[float] = None


@_dataclass
class ExecutionOptions:
    """Execution knobs (parity: execution_options.py ExecutionOptions);
    carried on DataContext / Train DataConfig."""

    resource_limits: ExecutionResources = _field(
        default_factory=ExecutionResources)
    exclude_resources: ExecutionResources = _field(
        default_factory=ExecutionResources)
    locality_with_output: bool = False
    preserve_order: bool = False
    actor_locality_enabled: bool = False
    verbose_progress: bool = False


@_dataclass
class TaskPoolStrategy:
    """Plain-task compute strategy marker (parity: reference
    TaskPoolStrategy); Data ops treat it the same as compute=None."""

    size: Optional[int] = None


@_dataclass
class FileShuffleConfig:
    """Seeded shuffle of input-file order for read_* APIs (parity:
    reference FileShuffleConfig)."""

    seed: Optional[int] = None


__all__ = [
    "ActorPoolStrategy", "BlockBasedFileDatasink", "DataContext",
    "DataIterator", "Dataset", "DatasetContext", "DatasetIterator",
    "DatasetSummary", "Datasink", "Datasource", "ExecutionOptions",
    "ExecutionResources", "FileShuffleConfig", "MaterializedDataset",
    "NodeIdStr", "Preprocessor", "ReadTask", "RowBasedFileDatasink",
    "SaveMode", "Schema", "SinkMode", "TaskPoolStrategy",
    "range", "range_tensor", "from_items",
    "from_numpy", "from_numpy_refs", "from_pandas", "from_pandas_refs",
    "from_arrow", "from_arrow_refs", "from_torch",
    "from_huggingface", "read_parquet", "read_csv", "read_datasource",
    "read_json", "read_sql", "read_text",
    "read_binary_files", "read_numpy", "col", "lit",
]

from builtins import range as _builtin_range  # noqa: E402 (public `range` below shadows it)


def _file_read_tasks(paths, reader, suffixes=None):
    files: List[str] = []
    for p in ([paths] if isinstance(paths, str) else paths):
        if os.path.isdir(p):
            for f in sorted(_glob.glob(os.path.join(p, "**", "*"), recursive=True)):
                if os.path.isfile(f) and (
                        not suffixes or any(f.endswith(s) for s in suffixes)):
                    files.append(f)
        else:
            files.extend(sorted(_glob.glob(p)) or [p])
    if not files:
        raise FileNotFoundError(f"no input files for {paths}")
    return [lambda f=f: reader(f) for f in files]


def range(n: int, *, parallelism: int = -1, override_num_blocks=None) -> Dataset:
    blocks = override_num_blocks or (parallelism if parallelism > 0 else
                                     min(max(1, n // 1000), 64))
    per = -(-n // blocks) if n else 1

    def make(start, end):
        def read():
            return {"id": np.arange(start, end, dtype=np.int64)}

        return read

    tasks = [make(s, min(s + per, n)) for s in _builtin_range(0, max(n, 1), per)
             if s < n or n == 0]
    if n == 0:
        tasks = [lambda: {"id": np.array([], dtype=np.int64)}]
    return Dataset([ReadOp(name=f"Range[{n}]", read_tasks=tasks, num_rows=n)])


def range_tensor(n: int, *, shape=(1,), parallelism: int = -1,
                 override_num_blocks=None) -> Dataset:
    blocks = override_num_blocks or (parallelism if parallelism > 0 else
                                     min(max(1, n // 1000), 64))
    per = -(-n // blocks) if n else 1

    def make(start, end):
        def read():
            count = end - start
            data = np.broadcast_to(
                np.arange(start, end, dtype=np.int64).reshape(
                    (count,) + (1,) * len(shape)),
                (count,) + tuple(shape),
            ).copy()
            return {"data": data}

        return read

    tasks = [make(s, min(s + per, n)) for s in _builtin_range(0, max(n, 1), per)
             if s < n]
    return Dataset([ReadOp(name=f"RangeTensor[{n}]", read_tasks=tasks, num_rows=n)])


def from_items(items: List[Any], *, parallelism: int = -1,
               override_num_blocks=None) -> Dataset:
    items = list(items)
    blocks = override_num_blocks or (parallelism if parallelism > 0 else
                                     min(max(1, len(items) // 100), 16))
    per = -(-len(items) // blocks) if items else 1

    def make(chunk):
        def read():
            rows = [it if isinstance(it, dict) else {"item": it} for it in chunk]
            return rows

        return read

    chunks = [items[s:s + per] for s in _builtin_range(0, len(items), per)] or [[]]
    return Dataset([ReadOp(name="FromItems", read_tasks=[make(c) for c in chunks],
                           num_rows=len(items))])


def _connected() -> bool:
    from ant_ray_amd._private.worker import global_worker

    return bool(getattr(global_worker, "connected", False))


def from_numpy(arr: np.ndarray, column: str = "data") -> Dataset:
    """Blocks are converted to arrow and put into the object store HERE
    (reference from_numpy eagerly stores blocks too): downstream stages
    receive shm refs — no closure-captured 200 MB array pickled per read
    task, no worker round trip re-storing the same bytes. Large inputs
    split into ~64 MB blocks for pipeline parallelism. Before ray.init
    the source stays lazy (read tasks) for compatibility."""
    import builtins

    from ant_ray_amd.data.block import BlockAccessor

    n = len(arr)
    if n and _connected():
        import ant_ray_amd as ray

        per = max(1, (64 << 20) // max(arr.nbytes // n, 1))
        refs = [
            ray.put(
                BlockAccessor.for_block({column: arr[s : s + per]}).to_arrow())
            for s in builtins.range(0, n, per)  # data.range shadows builtins
        ]
        return Dataset([ReadOp(name="FromNumpy", read_tasks=[], num_rows=n,
                               block_refs=refs)])

    def read():
        return {column: arr}

    return Dataset([ReadOp(name="FromNumpy", read_tasks=[read] if n else [],
                           num_rows=n)])


def from_pandas(dfs) -> Dataset:
    import pyarrow as pa

    dfs = [dfs] if not isinstance(dfs, list) else dfs
    n = sum(len(d) for d in dfs)
    if _connected():
        import ant_ray_amd as ray

        refs = [ray.put(pa.Table.from_pandas(d, preserve_index=False))
                for d in dfs]
        return Dataset([ReadOp(name="FromPandas", read_tasks=[],
                               num_rows=n, block_refs=refs)])
    tasks = [lambda d=d: pa.Table.from_pandas(d, preserve_index=False)
             for d in dfs]
    return Dataset([ReadOp(name="FromPandas", read_tasks=tasks, num_rows=n)])


def from_arrow(tables) -> Dataset:
    tables = [tables] if not isinstance(tables, list) else tables
    n = sum(t.num_rows for t in tables)
    if _connected():
        import ant_ray_amd as ray

        # driver-resident tables: put them (zero-copy for arrow buffers)
        # and skip the read-task round trip, as in from_numpy
        refs = [ray.put(t) for t in tables]
        return Dataset([ReadOp(name="FromArrow", read_tasks=[],
                               num_rows=n, block_refs=refs)])
    tasks = [lambda t=t: t for t in tables]
    return Dataset([ReadOp(name="FromArrow", read_tasks=tasks, num_rows=n)])


def from_torch(torch_dataset) -> Dataset:
    def read():
        rows = []
        for item in torch_dataset:
            rows.append({"item": item})
        return rows

    return Dataset([ReadOp(name="FromTorch", read_tasks=[read])])


def from_huggingface(hf_dataset) -> Dataset:
    try:
        table = hf_dataset.data.table  # datasets.Dataset holds an arrow table
        return from_arrow(table)
    except AttributeError:
        return from_items(list(hf_dataset))


def read_parquet(paths, *, columns=None, parallelism: int = -1,
                 filter_expr=None, **kw) -> Dataset:
    """Parquet source with native projection + filter pushdown: the
    optimizer folds downstream select_columns/filter_expr into the read
    tasks (pyarrow.parquet reads only the pruned columns/row groups —
    parity: reference logical/rules + ParquetDatasource pushdown)."""

    def make_read(cols, fexpr):
        def reader(f):
            import pyarrow.parquet as pq

            t = pq.read_table(f, columns=cols)
            if fexpr is not None:
                from ant_ray_amd.data.expressions import eval_expr_to_column

                t = t.filter(eval_expr_to_column(t, fexpr))
            return t

        return ReadOp(
            name="ReadParquet",
            read_tasks=_file_read_tasks(paths, reader, [".parquet"]),
            pushdown=lambda columns=None, filter_expr=None: make_read(
                sorted(set(cols or []) | set(columns)) if (cols and columns)
                else (columns or cols), fexpr if filter_expr is None
                else (filter_expr if fexpr is None else fexpr & filter_expr)),
        )

    return Dataset([make_read(columns, filter_expr)])


def read_csv(paths, *, parallelism: int = -1, **kw) -> Dataset:
    def reader(f):
        import pyarrow.csv as pcsv

        return pcsv.read_csv(f)

    return Dataset([ReadOp(name="ReadCSV",
                           read_tasks=_file_read_tasks(paths, reader, [".csv"]))])


def read_json(paths, *, parallelism: int = -1, **kw) -> Dataset:
    def reader(f):
        import pyarrow.json as pjson

        return pjson.read_json(f)

    return Dataset([ReadOp(name="ReadJSON",
                           read_tasks=_file_read_tasks(paths, reader,
                                                       [".json", ".jsonl"]))])


def read_text(paths, *, parallelism: int = -1, **kw) -> Dataset:
    def reader(f):
        with open(f) as fh:
            return {"text": np.array([ln.rstrip("\n") for ln in fh], dtype=object)}

    return Dataset([ReadOp(name="ReadText",
                           read_tasks=_file_read_tasks(paths, reader))])


def read_binary_files(paths, *, include_paths: bool = False,
                      parallelism: int = -1, **kw) -> Dataset:
    def reader(f):
        with open(f, "rb") as fh:
            data = fh.read()
        row = {"bytes": [data]}
        if include_paths:
            row["path"] = [f]
        return row

    return Dataset([ReadOp(name="ReadBinary",
                           read_tasks=_file_read_tasks(paths, reader))])


def read_numpy(paths, *, parallelism: int = -1, **kw) -> Dataset:
    def reader(f):
        return {"data": np.load(f)}

    return Dataset([ReadOp(name="ReadNumpy",
                           read_tasks=_file_read_tasks(paths, reader, [".npy"]))])


def read_datasource(datasource, *, parallelism: int = -1,
                    override_num_blocks=None, **read_args) -> Dataset:
    """Read via a custom Datasource (parity: reference read_datasource →
    datasource.get_read_tasks, one Ray task per ReadTask)."""
    p = override_num_blocks or (parallelism if parallelism > 0 else 8)
    tasks = datasource.get_read_tasks(p, **read_args) \
        if read_args else datasource.get_read_tasks(p)

    def _one(t):
        def run(t=t):
            blocks = list(t() if callable(t) else t)
            if len(blocks) == 1:
                return blocks[0]
            return blocks

        return run

    return Dataset([ReadOp(name=f"Read{datasource.get_name()}",
                           read_tasks=[_one(t) for t in tasks])])


def from_pandas_refs(dfs) -> Dataset:
    """Dataset from ObjectRefs holding pandas DataFrames (parity:
    reference from_pandas_refs); frames stay in the object store and are
    converted to blocks inside read tasks."""
    import ant_ray_amd as ray

    refs = [dfs] if not isinstance(dfs, list) else dfs

    def conv(ref):
        def run(ref=ref):
            import pyarrow as pa

            return pa.Table.from_pandas(ray.get(ref), preserve_index=False)

        return run

    return Dataset([ReadOp(name="FromPandasRefs",
                           read_tasks=[conv(r) for r in refs])])


def from_numpy_refs(arrs, column: str = "data") -> Dataset:
    import ant_ray_amd as ray

    refs = [arrs] if not isinstance(arrs, list) else arrs

    def conv(ref):
        def run(ref=ref):
            return {column: ray.get(ref)}

        return run

    return Dataset([ReadOp(name="FromNumpyRefs",
                           read_tasks=[conv(r) for r in refs])])


def from_arrow_refs(tables) -> Dataset:
    import ant_ray_amd as ray

    refs = [tables] if not isinstance(tables, list) else tables

    def conv(ref):
        def run(ref=ref):
            return ray.get(ref)

        return run

    return Dataset([ReadOp(name="FromArrowRefs",
                           read_tasks=[conv(r) for r in refs])])


def read_sql(sql: str, connection_factory, *, parallelism: int = -1,
             **_) -> Dataset:
    """Read the results of a SQL query via a DB-API connection factory
    (parity: reference read_sql; works offline with sqlite3)."""

    def run():
        conn = connection_factory()
        try:
            cur = conn.cursor()
            cur.execute(sql)
            cols = [d[0] for d in cur.description]
            rows = cur.fetchall()
        finally:
            conn.close()
        return {c: [r[i] for r in rows] for i, c in enumerate(cols)}

    return Dataset([ReadOp(name="ReadSQL", read_tasks=[run])])


def _unavailable_reader(name: str, needs: str):
    def fn(*_a, **_k):
        raise NotImplementedError(
            f"ray.data.{name} requires {needs}, which is not available in "
            "this air-gapped MI355X image (no network egress, package not "
            "installed). See PARITY.md 'Known reductions'.")

    fn.__name__ = name
    return fn


# cloud / third-party-backed readers and converters: explicit,
# informative stubs (the packages/services they need don't exist here)
read_images = _unavailable_reader("read_images", "an image decoder (PIL)")
read_audio = _unavailable_reader("read_audio", "an audio decoder")
read_videos = _unavailable_reader("read_videos", "a video decoder")
read_avro = _unavailable_reader("read_avro", "fastavro")
read_tfrecords = _unavailable_reader("read_tfrecords", "tensorflow protos")
read_webdataset = _unavailable_reader("read_webdataset", "webdataset")
read_clickhouse = _unavailable_reader("read_clickhouse", "clickhouse-connect")
read_delta = _unavailable_reader("read_delta", "deltalake")
read_delta_sharing_tables = _unavailable_reader(
    "read_delta_sharing_tables", "delta-sharing")
read_hudi = _unavailable_reader("read_hudi", "hudi")
read_iceberg = _unavailable_reader("read_iceberg", "pyiceberg")
read_kafka = _unavailable_reader("read_kafka", "kafka-python")
read_lance = _unavailable_reader("read_lance", "lance")
read_mcap = _unavailable_reader("read_mcap", "mcap")
read_mongo = _unavailable_reader("read_mongo", "pymongo")
read_snowflake = _unavailable_reader("read_snowflake", "snowflake-connector")
read_unity_catalog = _unavailable_reader("read_unity_catalog",
                                         "databricks APIs")
from_daft = _unavailable_reader("from_daft", "daft")
from_dask = _unavailable_reader("from_dask", "dask")
from_mars = _unavailable_reader("from_mars", "mars")
from_modin = _unavailable_reader("from_modin", "modin")
from_spark = _unavailable_reader("from_spark", "pyspark")
from_tf = _unavailable_reader("from_tf", "tensorflow")


class KafkaAuthConfig:
    """Stub config (parity name: reference read_kafka auth config);
    read_kafka itself is unavailable offline."""


class ClickHouseTableSettings:
    """Stub config (parity name: reference read_clickhouse settings)."""


class TFXReadOptions:
    """Stub config (parity name: reference read_tfrecords TFX options)."""
