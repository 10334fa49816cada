#!/bin/bash
set -x
cd "$(dirname "$0")/.."
mkdir -p gpurun_out profiles
timeout 600 python -m pytest tests/ -q -m gpu > gpurun_out/r02f_gputests.log 2>&1
echo "suite=$?"; tail -3 gpurun_out/r02f_gputests.log
timeout 420 python bench.py --steps 10 --warmup 6 2>&1 | grep -E "metric|hipGraph"
timeout 300 python scripts/bench_sampling.py 2>&1 | grep metric
export TMPDIR=/tmp
( cd /tmp && timeout 420 rocprofv3 --kernel-trace --stats -d /tmp/pf -o hf -- python /root/repo/bench.py --steps 6 --warmup 8 > /root/repo/gpurun_out/r02f_prof.log 2>&1 )
echo "prof=$?"
DB=$(ls /tmp/pf/*hf*.db 2>/dev/null | head -1)
[ -n "$DB" ] && python scripts/profile_summary.py "$DB" profiles/r02_prof_final.md 6 450 && head -22 profiles/r02_prof_final.md
python -c "import __graft_entry__; __graft_entry__.smoke()"
echo "smoke=$?"
echo DONE
