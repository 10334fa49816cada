#!/bin/bash
# One-gpurun-call hardware validation of every round-2 draft kernel.
# Round-2 session starts with:
#   /usr/local/graft/bin/gpurun --timeout 900 -- 'bash scripts/validate_drafts.sh 2>&1 | tee gpurun_out/drafts.log'
# Then iterate on whatever fails, exactly like the conv-fwd kernel went
# design -> validated -> beats-MIOpen in four short calls.
set -ex
cd "$(dirname "$0")/.."

# conv fwd v3: double-buffered staging (expect >= v2; guide says +40% in
# this occupancy regime if the pipeline holds)
DCR_NATIVE_CONV_V3=1 python -m pytest tests/test_ops_gpu.py -k fwd_v3 -x -q

# conv backward drafts (bwd-weight pixel-split, bwd-data tap loop)
DCR_NATIVE_CONV_BWD=1 python -m pytest tests/test_ops_gpu.py -k nhwc_bwd -x -q

# device-state AdamW + hipGraph capture + whole-train-step wiring
DCR_DEV_ADAMW=1 python -m pytest tests/test_ops_gpu.py -k adamw_dev -x -q
DCR_DEV_ADAMW=1 python -m pytest tests/test_train_gpu.py -k device_state -x -q

# attention v2: bit-exact masked-tail MFMA skip
DCR_ATTN_V2=1 python -m pytest tests/test_ops_gpu.py -k attn_fwd_v2 -x -q

# perf A/B (v3/v2 conv ratio, v2/v1 attention ratio per shape)
python scripts/bench_conv.py
python scripts/bench_attention.py

# bf16-GEMM kNN vs fp32 (expect ~2x)
python scripts/bench_search.py --repeat 2
python scripts/bench_search.py --repeat 2 --bf16

# whole-model A/B: each draft is also wired into the real model behind its
# env var, so the full bench can A/B them directly (run separately — each
# is a fresh process and pays the MIOpen find warmup):
#   DCR_ATTN_V2=1        python bench.py --steps 10 --warmup 6
#   DCR_NATIVE_CONV_V3=1 python bench.py --steps 10 --warmup 6
#   DCR_DEV_ADAMW=1      python bench.py --steps 10 --warmup 6
