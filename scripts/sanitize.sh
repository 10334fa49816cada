#!/usr/bin/env bash
# Kernel-test sanitizer pass: run the GPU op tests with
# serialized kernel execution + HSA debug so races / faulted kernels fail
# loudly at the offending launch instead of corrupting later state.
# Run on an MI355X box:  bash scripts/sanitize.sh
set -euo pipefail
cd "$(dirname "$0")/.."
export AMD_SERIALIZE_KERNEL=3       # wait after each kernel launch
export AMD_SERIALIZE_COPY=3
export HSA_ENABLE_DEBUG=1
export HIP_LAUNCH_BLOCKING=1
python -m pytest tests/test_ops_gpu.py -q -x "$@"
