#!/bin/bash
# Round-2 GPU call 2: GEMM validation+bench, conv-bwd bench, RCCL on HW,
# dev-AdamW and conv-bwd whole-step A/B.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
run() { local name="$1"; shift; echo "=== SECTION $name ==="; timeout 420 "$@" > "gpurun_out/r02c2_${name}.log" 2>&1; echo "=== $name exit=$? ==="; tail -30 "gpurun_out/r02c2_${name}.log"; }

run gemm_tests python -m pytest tests/test_ops_gpu.py -k "gemm or dcr_linear" -x -q
run bench_gemm python scripts/bench_gemm.py
run bench_conv python scripts/bench_conv.py
run rccl python -m pytest tests/test_rccl_gpu.py -x -q
DCR_DEV_ADAMW=1       run bench_devadamw python bench.py --steps 10 --warmup 6
DCR_NATIVE_CONV_BWD=1 run bench_convbwd  python bench.py --steps 10 --warmup 6
echo ALL DONE
