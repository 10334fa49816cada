#!/bin/bash
# Round-2 GPU call 3: full GPU suite, HEAD bench (grad-gather + hybrid
# wgrad + vec8 GN + gen attention), HEAD profile, sanitizer pass.
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out profiles
run() { local name="$1"; shift; echo "=== SECTION $name ==="; timeout 600 "$@" > "gpurun_out/r02c3_${name}.log" 2>&1; echo "=== $name exit=$? ==="; tail -25 "gpurun_out/r02c3_${name}.log"; }

run gputests python -m pytest tests/ -q -m gpu -x -rs
run bench_head python bench.py --steps 10 --warmup 6

# HEAD steady-state profile (VERDICT weak#3): kernel-trace only
export TMPDIR=/tmp
( cd /tmp && timeout 420 rocprofv3 --kernel-trace --stats -d /tmp/prof_head -o head -- python /root/repo/bench.py --steps 6 --warmup 10 > /root/repo/gpurun_out/r02c3_prof.log 2>&1 )
echo "prof exit=$?"
DB=$(ls /tmp/prof_head/*head*.db 2>/dev/null | head -1)
[ -n "$DB" ] && python scripts/profile_summary.py "$DB" gpurun_out/r02_prof_head.md 6 500 && cp gpurun_out/r02_prof_head.md profiles/r02_prof_head.md && head -40 profiles/r02_prof_head.md

run sanitize bash scripts/sanitize.sh
echo ALL DONE
