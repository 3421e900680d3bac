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
from typing import Any, Dict, List, Optional

import numpy as np

from ant_ray_amd.data.block import Block, BlockAccessor, block_from_dict
from ant_ray_amd.data.context import DataContext
from ant_ray_amd.data.dataset import Dataset, MaterializedDataset
from ant_ray_amd.data.iterator import DataIterator
from ant_ray_amd.data.expressions import col, lit
from ant_ray_amd.data.plan import ActorPoolStrategy, ReadOp

__all__ = [
    "ActorPoolStrategy", "DataContext", "DataIterator", "Dataset",
    "MaterializedDataset", "range", "range_tensor", "from_items",
    "from_numpy", "from_pandas", "from_arrow", "from_torch",
    "from_huggingface", "read_parquet", "read_csv", "read_json", "read_text",
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


def from_numpy(arr: np.ndarray, column: str = "data") -> Dataset:
    def read():
        return {column: arr}

    return Dataset([ReadOp(name="FromNumpy", read_tasks=[read], num_rows=len(arr))])


def from_pandas(dfs) -> Dataset:
    import pyarrow as pa

    dfs = [dfs] if not isinstance(dfs, list) else dfs
    tasks = [lambda d=d: pa.Table.from_pandas(d, preserve_index=False) for d in dfs]
    return Dataset([ReadOp(name="FromPandas", read_tasks=tasks,
                           num_rows=sum(len(d) for d in dfs))])


def from_arrow(tables) -> Dataset:
    tables = [tables] if not isinstance(tables, list) else tables
    tasks = [lambda t=t: t for t in tables]
    return Dataset([ReadOp(name="FromArrow", read_tasks=tasks,
                           num_rows=sum(t.num_rows for t in tables))])


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


def read_parquet(paths, *, columns=None, parallelism: int = -1, **kw) -> Dataset:
    def reader(f):
        import pyarrow.parquet as pq

        return pq.read_table(f, columns=columns)

    return Dataset([ReadOp(name="ReadParquet",
                           read_tasks=_file_read_tasks(paths, reader, [".parquet"]))])


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
