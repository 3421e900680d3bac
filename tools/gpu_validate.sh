#!/bin/bash
# One-call GPU validation for a fresh MI355X box (round-2 first action):
#   /usr/local/graft/bin/gpurun --timeout 2700 -- 'bash tools/gpu_validate.sh'
# Produces gpurun_out/val/: test results, bench lines, flash-attention A/B,
# rocprofv3 kernel stats. Copy keepers into profiles/ and commit.
set -x
mkdir -p gpurun_out/val
cd "$(dirname "$0")/.." || exit 1

# 1) full GPU test suite (NOT -x: see every failure)
timeout 900 python -m pytest tests -m gpu -q > gpurun_out/val/pytest_gpu.log 2>&1
echo "pytest rc=$?" >> gpurun_out/val/pytest_gpu.log
tail -5 gpurun_out/val/pytest_gpu.log

# 2) smoke
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" \
    > gpurun_out/val/smoke.log 2>&1

# 3) bench: SDPA default vs flash opt-in (the round-1 open question)
timeout 900 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/val/bench_sdpa.json 2> gpurun_out/val/bench_sdpa.err
ANTRAY_FLASH=1 timeout 900 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/val/bench_flash.json 2> gpurun_out/val/bench_flash.err
echo "== SDPA:";  tail -1 gpurun_out/val/bench_sdpa.json
echo "== FLASH:"; tail -1 gpurun_out/val/bench_flash.json

# 4) dkv kernel microbench (dK prefetch change vs r01 numbers)
timeout 600 python tools/bench_dkv.py > gpurun_out/val/bench_dkv.log 2>&1 || true

# 5) kernel stats of one bench step (stats-only: PMC needs its own run)
cd /tmp && export TMPDIR=/tmp && cd - >/dev/null
timeout 900 rocprofv3 --kernel-trace --stats -d gpurun_out/val/prof -- \
    python bench.py --steps 3 --warmup 1 \
    > gpurun_out/val/rocprof_bench.log 2>&1 || true
find gpurun_out/val/prof -name "*stats*" | head -3

echo DONE
