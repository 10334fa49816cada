#!/bin/bash
# Final exit gate at round-2 HEAD: full GPU suite + bench + smoke.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out
timeout 600 python -m pytest tests/ -q -m gpu > gpurun_out/r02gate_tests.log 2>&1
echo "suite=$?"; tail -2 gpurun_out/r02gate_tests.log
timeout 420 python bench.py --steps 10 --warmup 6 2>&1 | grep -E "metric|hipGraph"
python -c "import __graft_entry__; __graft_entry__.smoke()"
echo "smoke=$?"
echo DONE
