#!/bin/bash
# Bucket-size sweep for the 8-GPU all-reduce (run on an 8-GPU lease):
#   bash tools/sweep_allreduce.sh [NGPUS]
# Writes gpurun_out/arsweep/bucket<mb>.json and prints a summary table.
set -e
N=${1:-8}
mkdir -p gpurun_out/arsweep
for MB in 32 64 128 256; do
  echo "== bucket_mb=$MB =="
  timeout 1200 python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
    --master-addr 127.0.0.1 --master-port 29617 \
    bench.py --bare --gpus "$N" --steps 4 --warmup 2 --bucket-mb "$MB" \
    2>gpurun_out/arsweep/bucket$MB.err | tail -1 | tee gpurun_out/arsweep/bucket$MB.json
done
echo "== summary =="
for MB in 32 64 128 256; do
  V=$(python -c "import json;print(json.load(open('gpurun_out/arsweep/bucket$MB.json'))['value'])" 2>/dev/null || echo NA)
  echo "bucket_mb=$MB tokens/s=$V"
done
