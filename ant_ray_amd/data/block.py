"""Block layer: a block is a pyarrow.Table; BlockAccessor adapts formats.

Role parity: reference python/ray/data/block.py + _internal/arrow_block.py
(blocks are Arrow tables in plasma; accessor converts to pandas/numpy and
builds batches). Blocks move between tasks as ObjectRefs through our shm
object store (zero-copy arrow buffers via the serialization fast path).
"""
from __future__ import annotations

from typing import Any, Dict, Iterator, List, Optional

import numpy as np
import pyarrow as pa

Block = pa.Table


def block_from_rows(rows: List[Dict[str, Any]]) -> Block:
    """Rows (dicts) -> arrow table. Non-scalar values become object columns
    via arrow's python-object inference; tensors stay numpy."""
    if not rows:
        return pa.table({})
    cols: Dict[str, list] = {k: [] for k in rows[0]}
    for r in rows:
        for k in cols:
            cols[k].append(r.get(k))
    return block_from_dict(cols)


def block_from_dict(cols: Dict[str, Any]) -> Block:
    arrays = {}
    for k, v in cols.items():
        if isinstance(v, np.ndarray) and v.ndim > 1:
            # tensor column: fixed-shape list array of flattened rows
            arrays[k] = _tensor_to_arrow(v)
        else:
            try:
                arrays[k] = pa.array(v)
            except (pa.ArrowInvalid, pa.ArrowNotImplementedError, pa.ArrowTypeError):
                arrays[k] = pa.array([_pickle_obj(o) for o in v], type=pa.binary())
    return pa.table(arrays)


_TENSOR_META = b"__antray_tensor__"


def _tensor_to_arrow(arr: np.ndarray) -> pa.Array:
    flat = arr.reshape(len(arr), -1)
    la = pa.FixedSizeListArray.from_arrays(pa.array(flat.ravel()), flat.shape[1])
    return la


def _pickle_obj(o):
    import pickle

    return _TENSOR_META + pickle.dumps(o)


def _maybe_unpickle(v):
    if isinstance(v, bytes) and v.startswith(_TENSOR_META):
        import pickle

        return pickle.loads(v[len(_TENSOR_META):])
    return v


class BlockAccessor:
    def __init__(self, block: Block):
        self.block = block

    @staticmethod
    def for_block(block) -> "BlockAccessor":
        if isinstance(block, pa.Table):
            return BlockAccessor(block)
        if isinstance(block, dict):
            return BlockAccessor(block_from_dict(block))
        if isinstance(block, list):
            return BlockAccessor(block_from_rows(block))
        try:
            import pandas as pd

            if isinstance(block, pd.DataFrame):
                return BlockAccessor(pa.Table.from_pandas(block, preserve_index=False))
        except ImportError:
            pass
        raise TypeError(f"cannot treat {type(block)} as a block")

    def num_rows(self) -> int:
        return self.block.num_rows

    def size_bytes(self) -> int:
        return self.block.nbytes

    def schema(self):
        return self.block.schema

    def slice(self, start: int, end: int) -> Block:
        return self.block.slice(start, end - start)

    def to_arrow(self) -> pa.Table:
        return self.block

    def to_pandas(self):
        return self.block.to_pandas()

    def to_numpy(self) -> Dict[str, np.ndarray]:
        out = {}
        for name in self.block.column_names:
            col = self.block.column(name)
            if pa.types.is_fixed_size_list(col.type):
                flat = col.combine_chunks().flatten().to_numpy(zero_copy_only=False)
                out[name] = flat.reshape(len(col), -1)
            else:
                v = col.to_numpy(zero_copy_only=False)
                if len(v) and isinstance(v[0], bytes) and v[0].startswith(_TENSOR_META):
                    v = np.array([_maybe_unpickle(x) for x in v], dtype=object)
                out[name] = v
        return out

    def iter_rows(self) -> Iterator[Dict[str, Any]]:
        for batch in self.block.to_batches():
            cols = batch.to_pydict()
            names = list(cols)
            for i in range(batch.num_rows):
                yield {k: _maybe_unpickle(cols[k][i]) for k in names}

    def to_batch(self, batch_format: Optional[str]):
        if batch_format in (None, "default", "numpy"):
            return self.to_numpy()
        if batch_format == "pandas":
            return self.to_pandas()
        if batch_format in ("pyarrow", "arrow"):
            return self.block
        raise ValueError(f"unknown batch_format {batch_format}")


def batch_to_block(batch) -> Block:
    if isinstance(batch, pa.Table):
        return batch
    if isinstance(batch, dict):
        return block_from_dict(batch)
    try:
        import pandas as pd

        if isinstance(batch, pd.DataFrame):
            return pa.Table.from_pandas(batch, preserve_index=False)
    except ImportError:
        pass
    raise TypeError(
        f"map_batches UDF must return dict/pandas/pyarrow, got {type(batch)}"
    )
