"""Summarize a rocprofv3 --pmc run: per-kernel counter totals.

Usage: python tools/pmc_summary.py <rocprof -d dir> [sort_counter]

Reads every *counter_collection.csv under the directory, sums counter
values per (kernel, counter), and prints one row per kernel with the
counters as columns, sorted by sort_counter (default GRBM_GUI_ACTIVE).
Derived columns when the inputs are present:
  * mfma_ratio = SQ_VALU_MFMA_BUSY_CYCLES / SQ_BUSY_CYCLES — relative
    matrix-core pressure (the MFMA counter aggregates per-pipe passes, so
    this is a RELATIVE ranking across kernels, not an absolute percent)
  * hbm_gb     = (FETCH_SIZE + WRITE_SIZE) in GB
"""
import glob
import os
import sys

import pandas as pd


def main():
    root = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/pmc"
    sort_key = sys.argv[2] if len(sys.argv) > 2 else None
    files = glob.glob(os.path.join(root, "**", "*counter_collection.csv"),
                      recursive=True)
    if not files:
        print(f"no counter_collection.csv under {root}")
        return 1
    df = pd.concat([pd.read_csv(f) for f in files], ignore_index=True)
    kcol = next(c for c in df.columns if "Kernel_Name" in c or c == "Kernel_Name")
    ccol = next(c for c in df.columns if "Counter_Name" in c)
    vcol = next(c for c in df.columns if "Counter_Value" in c)
    df[kcol] = df[kcol].str.replace(r"\(.*\)", "", regex=True).str.split("<").str[0].str.strip()
    pivot = df.pivot_table(index=kcol, columns=ccol, values=vcol,
                           aggfunc="sum")
    ndisp = df.groupby(kcol)["Dispatch_Id"].nunique() if "Dispatch_Id" in df.columns else None
    if ndisp is not None:
        pivot["dispatches"] = ndisp
    if {"SQ_VALU_MFMA_BUSY_CYCLES", "SQ_BUSY_CYCLES"} <= set(pivot.columns):
        pivot["mfma_ratio"] = (pivot["SQ_VALU_MFMA_BUSY_CYCLES"]
                               / pivot["SQ_BUSY_CYCLES"]).round(2)
    if {"SQ_LDS_BANK_CONFLICT", "SQ_BUSY_CYCLES"} <= set(pivot.columns):
        pivot["lds_conf_frac"] = (pivot["SQ_LDS_BANK_CONFLICT"]
                                  / pivot["SQ_BUSY_CYCLES"]).round(3)
    if {"FETCH_SIZE", "WRITE_SIZE"} <= set(pivot.columns):
        pivot["hbm_gb"] = ((pivot["FETCH_SIZE"] + pivot["WRITE_SIZE"])
                           / 1024).round(2)
    if sort_key is None:
        sort_key = ("GRBM_GUI_ACTIVE" if "GRBM_GUI_ACTIVE" in pivot.columns
                    else pivot.columns[0])
    pivot = pivot.sort_values(sort_key, ascending=False)
    pd.set_option("display.width", 200)
    pd.set_option("display.max_colwidth", 60)
    print(pivot.head(20).to_string())
    return 0


if __name__ == "__main__":
    sys.exit(main())
