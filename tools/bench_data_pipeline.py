"""BASELINE config 5 measurement (1-GPU slice): Ray Data streaming
pipeline throughput into a GPU consumer — u8 batches cross H2D as bytes
and cast on-device via the fused data_transform kernels.

Run on the GPU box: python tools/bench_data_pipeline.py
Writes profiles-ready JSON to stdout.
"""
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    import torch

    import ant_ray_amd as ray
    import ant_ray_amd.data as data

    ray.init(num_cpus=8, num_gpus=1)
    n_rows, dim = 200_000, 1024  # ~200 MB u8
    raw = (np.random.rand(n_rows, dim) * 255).astype(np.uint8)
    ds = data.from_numpy(raw).map_batches(
        lambda b: {"data": b["data"]}, batch_size=4096)
    # warm one pass
    it = ds.iter_torch_batches(batch_size=4096,
                               dtypes={"data": torch.bfloat16},
                               device="cuda")
    rows = 0
    t0 = time.time()
    for batch in it:
        x = batch["data"]
        assert x.is_cuda and x.dtype == torch.bfloat16
        rows += len(x)
    torch.cuda.synchronize()
    wall = time.time() - t0
    out = {
        "metric": "Data streaming GPU-collate MB/s (u8->bf16 on-device)",
        "rows": rows, "dim": dim,
        "mb_per_s": round(rows * dim / wall / 1e6, 1),
        "rows_per_s": round(rows / wall, 1),
        "wall_s": round(wall, 2),
    }
    print(json.dumps(out), flush=True)
    ray.shutdown()


if __name__ == "__main__":
    main()
