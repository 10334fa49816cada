#!/bin/bash
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
timeout 240 python -m pytest tests/test_ops_gpu.py -k "attn or attention" -q -x > gpurun_out/r02c8_attn_tests.log 2>&1
echo "tests=$?"; tail -3 gpurun_out/r02c8_attn_tests.log
timeout 300 python scripts/bench_sampling.py 2>&1 | grep metric
timeout 420 python bench.py --steps 10 --warmup 6 2>&1 | grep metric
timeout 420 python scripts/profile_aten.py > gpurun_out/r02c8_aten.log 2>&1
echo "aten=$?"; grep -A40 "Name.*Self CPU" gpurun_out/r02c8_aten.log | head -10
echo DONE
