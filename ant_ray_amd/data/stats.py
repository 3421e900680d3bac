"""Dataset summary statistics.

Role parity: reference python/ray/data/stats.py DatasetSummary (per-column
count/min/max/mean/std/missing-percentage), computed distributed: one Ray
task per block emits partial moments, the driver merges them.
"""
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


def _block_partials(block, cols):
    """Per-column partial stats for one block (runs inside a Ray task)."""
    import numpy as np

    from ant_ray_amd.data.block import BlockAccessor

    acc = BlockAccessor(block)
    out = {}
    tbl = acc.to_arrow()
    for c in cols:
        if c not in tbl.column_names:
            continue
        arr = tbl.column(c).to_numpy(zero_copy_only=False)
        n = len(arr)
        if arr.dtype == object:
            missing = sum(1 for v in arr if v is None)
            out[c] = {"n": n, "missing": missing}
            continue
        mask = np.isnan(arr.astype("float64", copy=False)) \
            if np.issubdtype(arr.dtype, np.floating) else np.zeros(n, bool)
        vals = arr[~mask].astype("float64", copy=False)
        out[c] = {
            "n": n, "missing": int(mask.sum()),
            "min": float(vals.min()) if len(vals) else None,
            "max": float(vals.max()) if len(vals) else None,
            "sum": float(vals.sum()), "sumsq": float((vals ** 2).sum()),
            "cnt": int(len(vals)),
        }
    return out


def _summarize_blocks(ds, cols) -> Dict[str, Dict[str, Any]]:
    import ant_ray_amd as ray

    task = ray.remote(_block_partials)
    partials = ray.get([task.remote(ref, cols)
                        for ref in ds.iter_internal_ref_bundles()])
    merged: Dict[str, Dict[str, Any]] = {}
    for part in partials:
        for c, p in part.items():
            m = merged.setdefault(c, {"n": 0, "missing": 0, "cnt": 0,
                                      "sum": 0.0, "sumsq": 0.0,
                                      "min": None, "max": None})
            m["n"] += p["n"]
            m["missing"] += p["missing"]
            if "cnt" in p:
                m["cnt"] += p["cnt"]
                m["sum"] += p["sum"]
                m["sumsq"] += p["sumsq"]
                for k, f in (("min", min), ("max", max)):
                    if p[k] is not None:
                        m[k] = p[k] if m[k] is None else f(m[k], p[k])
    out = {}
    for c, m in merged.items():
        row = {"count": m["n"], "missing_pct": 100.0 * m["missing"] / m["n"]
               if m["n"] else 0.0}
        if m["cnt"]:
            mean = m["sum"] / m["cnt"]
            var = max(m["sumsq"] / m["cnt"] - mean * mean, 0.0)
            row.update(min=m["min"], max=m["max"], mean=mean,
                       std=var ** 0.5)
        out[c] = row
    return out


@dataclass
class DatasetSummary:
    """Computed statistics per column; to_pandas() gives the stats table
    (parity: reference DatasetSummary.to_pandas)."""

    dataset_schema: Any
    columns: List[str]
    stats: Dict[str, Dict[str, Any]] = field(default_factory=dict)

    STATISTIC_COLUMN = "statistic"

    def column_stats(self, column: str) -> Dict[str, Any]:
        return self.stats.get(column, {})

    def to_pandas(self):
        import pandas as pd

        names = sorted({k for v in self.stats.values() for k in v})
        rows = []
        for stat in names:
            row = {self.STATISTIC_COLUMN: stat}
            for c in self.columns:
                row[c] = self.stats.get(c, {}).get(stat)
            rows.append(row)
        return pd.DataFrame(rows)
