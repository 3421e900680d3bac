"""Custom datasources and datasinks.

Role parity: reference python/ray/data/datasource/datasource.py
(Datasource / ReadTask), datasink.py (Datasink, on_write_start/write/
on_write_complete lifecycle), file_datasink.py (RowBasedFileDatasink /
BlockBasedFileDatasink) and SaveMode/SinkMode. Reads fan out one Ray task
per ReadTask; writes fan out one task per block ref-bundle, with the
driver running the start/complete hooks.
"""
from __future__ import annotations

import enum
import os
from typing import Any, Callable, Iterable, List, Optional


class ReadTask:
    """One unit of read parallelism: a no-arg callable producing one or
    more blocks (dict-of-columns / pandas / pyarrow), plus optional
    metadata used for planning."""

    def __init__(self, read_fn: Callable[[], Iterable[Any]],
                 metadata: Optional[dict] = None):
        self._read_fn = read_fn
        self.metadata = metadata or {}

    def __call__(self):
        return self._read_fn()


class Datasource:
    """Subclass and implement get_read_tasks(parallelism) -> List[ReadTask]
    (and optionally estimate_inmemory_data_size)."""

    def get_name(self) -> str:
        return type(self).__name__.replace("Datasource", "")

    def estimate_inmemory_data_size(self) -> Optional[int]:
        return None

    def get_read_tasks(self, parallelism: int) -> List[ReadTask]:
        raise NotImplementedError


class SaveMode(enum.Enum):
    APPEND = "append"
    OVERWRITE = "overwrite"
    IGNORE = "ignore"
    ERROR = "error"


# alias used by some reference call sites
SinkMode = SaveMode


class Datasink:
    """Subclass and implement write(); the driver calls on_write_start
    first, then one write() per block bundle inside Ray tasks, then
    on_write_complete(results) (or on_write_failed)."""

    def on_write_start(self) -> None:
        pass

    def write(self, blocks: Iterable[Any], ctx: Optional[dict] = None) -> Any:
        raise NotImplementedError

    def on_write_complete(self, write_result_blocks: List[Any]) -> None:
        pass

    def on_write_failed(self, error: Exception) -> None:
        pass

    @property
    def supports_distributed_writes(self) -> bool:
        return True

    def get_name(self) -> str:
        return type(self).__name__.replace("Datasink", "")


class _FileDatasink(Datasink):
    def __init__(self, path: str, *, file_format: str = "out",
                 mode: SaveMode = SaveMode.APPEND, **_):
        self.path = path
        self.file_format = file_format
        self.mode = mode if isinstance(mode, SaveMode) else SaveMode(mode)

    def on_write_start(self):
        if os.path.isdir(self.path) and os.listdir(self.path):
            if self.mode == SaveMode.ERROR:
                raise ValueError(
                    f"output path {self.path!r} is not empty (SaveMode.ERROR)")
            if self.mode == SaveMode.OVERWRITE:
                for f in os.listdir(self.path):
                    fp = os.path.join(self.path, f)
                    if os.path.isfile(fp):
                        os.unlink(fp)
        os.makedirs(self.path, exist_ok=True)

    def _open(self, task_index: int):
        name = f"{task_index:06d}_{os.getpid()}.{self.file_format}"
        return open(os.path.join(self.path, name), "wb")


class RowBasedFileDatasink(_FileDatasink):
    """Implement write_row_to_file(row, file) — one output file per block
    task, rows streamed through it (parity: file_datasink.py)."""

    def write_row_to_file(self, row: dict, file) -> None:
        raise NotImplementedError

    def write(self, blocks, ctx=None):
        from ant_ray_amd.data.block import BlockAccessor

        idx = (ctx or {}).get("task_index", 0)
        n = 0
        if self.mode == SaveMode.IGNORE and os.listdir(self.path):
            return 0
        with self._open(idx) as f:
            for b in blocks:
                for row in BlockAccessor(b).iter_rows():
                    self.write_row_to_file(row, f)
                    n += 1
        return n


class BlockBasedFileDatasink(_FileDatasink):
    """Implement write_block_to_file(block, file) — one file per block
    task (parity: file_datasink.py)."""

    def write_block_to_file(self, block, file) -> None:
        raise NotImplementedError

    def write(self, blocks, ctx=None):
        from ant_ray_amd.data.block import BlockAccessor

        idx = (ctx or {}).get("task_index", 0)
        n = 0
        if self.mode == SaveMode.IGNORE and os.listdir(self.path):
            return 0
        with self._open(idx) as f:
            for b in blocks:
                self.write_block_to_file(BlockAccessor(b).to_arrow(), f)
                n += BlockAccessor(b).num_rows()
        return n
