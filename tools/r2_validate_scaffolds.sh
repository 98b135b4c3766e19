#!/usr/bin/env bash
# Round-2 opener: validate + A/B the dormant kernel scaffolds on a GPU box.
# Run via:  /usr/local/graft/bin/gpurun --timeout 900 -- 'bash tools/r2_validate_scaffolds.sh'
# Budget: ~6-8 GPU-minutes. Results land in gpurun_out/.
set -euo pipefail
mkdir -p gpurun_out

echo "== full GPU suite (incl. new pointwise/unet) =="
timeout 600 python -m pytest tests/ -q -m gpu 2>&1 \
    | tee gpurun_out/r2_gpu_suite.log | tail -5 || true

echo "== gated numerics (CTILE=1, fwd-DB, wgrad-DB) =="
COINN_SPATIAL_CI1=1 COINN_SPATIAL_DB=1 COINN_WGRAD_DB=1 \
    timeout 420 python -m pytest tests/test_gpu_conv3d.py -q \
    -k "ctile1 or double_buffered" 2>&1 \
    | tee gpurun_out/r2_scaffold_tests.log || true

echo "== A/B bench: baseline =="
timeout 180 python bench.py --steps 15 --warmup 5 \
    > gpurun_out/r2_bench_baseline.json 2>gpurun_out/r2_bench_baseline.log

echo "== A/B bench: CTILE=1 routing on =="
COINN_SPATIAL_CI1=1 timeout 180 python bench.py --steps 15 --warmup 5 \
    > gpurun_out/r2_bench_ci1.json 2>/dev/null

echo "== A/B bench: fwd double-buffer on =="
COINN_SPATIAL_DB=1 timeout 180 python bench.py --steps 15 --warmup 5 \
    > gpurun_out/r2_bench_fwddb.json 2>/dev/null

echo "== A/B bench: wgrad double-buffer on =="
COINN_WGRAD_DB=1 timeout 180 python bench.py --steps 15 --warmup 5 \
    > gpurun_out/r2_bench_wgraddb.json 2>/dev/null

echo "== A/B bench: all on =="
COINN_SPATIAL_CI1=1 COINN_SPATIAL_DB=1 COINN_WGRAD_DB=1 \
    timeout 180 python bench.py --steps 15 --warmup 5 \
    > gpurun_out/r2_bench_all.json 2>/dev/null

echo "== summary =="
for f in gpurun_out/r2_bench_*.json; do
    echo "$f: $(tail -1 "$f")"
done
