#!/bin/bash
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
# tiny-model graph test with full traceback
timeout 240 python -m pytest tests/test_train_gpu.py::test_hipgraph_captured_step -q -x --tb=long > gpurun_out/r02c11_graph.log 2>&1
echo "graphtest=$?"; tail -30 gpurun_out/r02c11_graph.log
if grep -q "1 passed" gpurun_out/r02c11_graph.log; then
  DCR_HIPGRAPH=1 timeout 420 python bench.py --steps 10 --warmup 6 2>&1 | grep -E "metric|Error" | head -4
fi
echo DONE
