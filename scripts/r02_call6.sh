#!/bin/bash
# HEAD profile + sampling bench + final bench (script file: inline $() in
# gpurun commands gets expanded by the outer shell — calls 4/5 lost their
# profile summaries to that).
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out profiles
export TMPDIR=/tmp
( cd /tmp && timeout 420 rocprofv3 --kernel-trace --stats -d /tmp/p6 -o h6 -- python /root/repo/bench.py --steps 6 --warmup 8 > /root/repo/gpurun_out/r02c6_prof.log 2>&1 )
echo "prof=$?"
DB=$(ls /tmp/p6/*h6*.db 2>/dev/null | head -1)
echo "DB=$DB"
if [ -n "$DB" ]; then
  python scripts/profile_summary.py "$DB" profiles/r02_prof_head.md 6 450
  head -30 profiles/r02_prof_head.md
fi
timeout 300 python scripts/bench_sampling.py > gpurun_out/r02c6_sampling.log 2>&1
echo "sampling=$?"; grep -v libdrm gpurun_out/r02c6_sampling.log | tail -6
timeout 420 python bench.py --steps 10 --warmup 6 2>&1 | grep metric
echo DONE
