#!/bin/bash
# One-call GPU validation for a fresh MI355X box (round-2 first action):
#   /usr/local/graft/bin/gpurun --timeout 2100 -- 'bash tools/gpu_validate.sh'
# Produces gpurun_out/val/: test results, bench lines (via-ray + bare),
# flash-attention A/B. Copy keepers into profiles/ and commit.
set -x
mkdir -p gpurun_out/val
cd "$(dirname "$0")/.." || exit 1

# 1) full GPU test suite (NOT -x: see every failure) — the r01 headline item
timeout 1000 python -m pytest tests -m gpu -q > gpurun_out/val/pytest_gpu.log 2>&1
echo "pytest rc=$?" >> gpurun_out/val/pytest_gpu.log
tail -5 gpurun_out/val/pytest_gpu.log

# 2) smoke
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" \
    > gpurun_out/val/smoke.log 2>&1

# 3) bench --bare: SDPA default vs flash opt-in (isolates the kernels)
timeout 600 python bench.py --bare --steps 6 --warmup 2 \
    > gpurun_out/val/bench_bare_sdpa.json 2> gpurun_out/val/bench_bare_sdpa.err
ANTRAY_FLASH=1 timeout 600 python bench.py --bare --steps 6 --warmup 2 \
    > gpurun_out/val/bench_bare_flash.json 2> gpurun_out/val/bench_bare_flash.err
echo "== BARE SDPA:";  tail -1 gpurun_out/val/bench_bare_sdpa.json
echo "== BARE FLASH:"; tail -1 gpurun_out/val/bench_bare_flash.json

# 4) bench default = via Ray Train (TorchTrainer + PG): the reported path
timeout 900 python bench.py --steps 6 --warmup 2 \
    > gpurun_out/val/bench_viaray.json 2> gpurun_out/val/bench_viaray.err
echo "== VIA-RAY:"; tail -1 gpurun_out/val/bench_viaray.json
tail -3 gpurun_out/val/bench_viaray.err

# 5) dkv kernel microbench (dK prefetch change vs r01 numbers) + attn A/B
timeout 600 python tools/bench_dkv.py > gpurun_out/val/bench_dkv.log 2>&1 || true
timeout 600 python tools/bench_attn.py > gpurun_out/val/bench_attn.log 2>&1 || true
tail -12 gpurun_out/val/bench_attn.log

echo DONE
