#!/bin/bash
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
# A/B: graph-captured step vs eager at the same HEAD (also picks up the
# clip-cast and arena-zero-skip wins)
timeout 420 python bench.py --steps 10 --warmup 6 2>&1 | grep metric
DCR_HIPGRAPH=1 timeout 420 python bench.py --steps 10 --warmup 6 2>&1 | grep -E "metric|Error|error" | head -5
echo DONE
