"""Ray Data: streaming pipeline with expressions + GPU stage -> Train shards.

    python examples/data_pipeline.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ray
from ray.data import col

import ant_ray_amd.data as data

if __name__ == "__main__":
    ray.init()
    ds = (data.range(10_000, override_num_blocks=32)
          .with_columns({"sq": col("id") * col("id")})
          .filter_expr(col("sq") % 2 == 0)
          .map_batches(lambda b: {"id": b["id"], "sq": b["sq"],
                                  "norm": b["sq"] / b["sq"].max()}))
    print("count:", ds.count())
    print(ds.take(3))
    shards = ds.streaming_split(2, equal=True)
    for i, batch in enumerate(shards[0].iter_torch_batches(batch_size=256)):
        if i == 0:
            print("shard batch:", {k: v.shape for k, v in batch.items()})
    ray.shutdown()
