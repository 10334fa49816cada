#!/bin/bash
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
timeout 240 python -m pytest tests/test_ops_gpu.py -k "attn_fwd_v4 or flash_attention" -q -x > gpurun_out/r02c7_v4.log 2>&1
echo "v4tests=$?"; tail -4 gpurun_out/r02c7_v4.log
timeout 240 python scripts/bench_attention.py 2>&1 | grep -v libdrm
echo DONE
