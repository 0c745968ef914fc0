#!/bin/bash
# Round-1 GPU validation: wave-solver numerics + microbench + 1-GPU bench + rocprof
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== gpu tests ==="
timeout 300 python -m pytest tests -m gpu -x -q 2>&1 | tail -4
echo "=== solver microbench (wave kernel) ==="
timeout 400 python scripts/solver_microbench.py 2000000
echo "=== bench 1-GPU default (north-star shard) ==="
timeout 900 python bench.py --steps 3 --warmup 1 2> gpurun_out/bench1.log | tee gpurun_out/bench1.json
tail -3 gpurun_out/bench1.log
echo "=== rocprof kernel stats (small bench) ==="
export TMPDIR=/tmp
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof -o als -- \
  python /root/repo/bench.py --steps 2 --warmup 1 --users-per-gpu 2000000 --items 2000000 \
  > /root/repo/gpurun_out/bench_prof.json 2> /root/repo/gpurun_out/prof_bench.log || tail -20 /root/repo/gpurun_out/prof_bench.log
ls -la /root/repo/gpurun_out/prof/ 2>/dev/null
for f in /root/repo/gpurun_out/prof/*kernel_stats*; do echo "--- $f"; head -15 "$f"; done
