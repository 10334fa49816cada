#!/bin/bash
# Round-2 opening lease: validate every compiled draft + HEAD bench, all in
# one gpurun call. Failure-tolerant: each section runs regardless of earlier
# failures so the whole A/B picture comes back from a single lease.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out

run() {  # run <name> <cmd...>
  local name="$1"; shift
  echo "=== SECTION $name ==="
  timeout 300 "$@" > "gpurun_out/r02_${name}.log" 2>&1
  echo "=== SECTION $name exit=$? ==="
  tail -25 "gpurun_out/r02_${name}.log"
}

# HEAD bench first (VERDICT weak#1: reconcile 185 vs 198)
run bench_head python bench.py --steps 10 --warmup 6

# draft validations (env-gated tests)
DCR_NATIVE_CONV_V3=1  run conv_v3   python -m pytest tests/test_ops_gpu.py -k fwd_v3 -x -q
DCR_NATIVE_CONV_BWD=1 run conv_bwd  python -m pytest tests/test_ops_gpu.py -k nhwc_bwd -x -q
DCR_DEV_ADAMW=1       run adamw_dev python -m pytest tests/test_ops_gpu.py -k adamw_dev -x -q
DCR_DEV_ADAMW=1       run adamw_train python -m pytest tests/test_train_gpu.py -k device_state -x -q
DCR_ATTN_V2=1         run attn_v2   python -m pytest tests/test_ops_gpu.py -k attn_fwd_v2 -x -q

# perf A/B tables
run bench_conv python scripts/bench_conv.py
run bench_attn python scripts/bench_attention.py
run knn_fp32 python scripts/bench_search.py --repeat 2
run knn_bf16 python scripts/bench_search.py --repeat 2 --bf16

echo "ALL SECTIONS DONE"
