#!/bin/bash
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
# conv dispatch + train-step integrity with the layout-preserving flatten
timeout 300 python -m pytest tests/test_train_gpu.py tests/test_ops_gpu.py -k "train or conv or adamw or resnet" -q -x > gpurun_out/r02c9_tests.log 2>&1
echo "tests=$?"; tail -3 gpurun_out/r02c9_tests.log
timeout 420 python bench.py --steps 10 --warmup 6 2>&1 | grep metric
timeout 360 python scripts/profile_aten.py > gpurun_out/r02c9_aten.log 2>&1
echo "aten=$?"; grep -E "copy_|conv" gpurun_out/r02c9_aten.log | head -8
echo DONE
