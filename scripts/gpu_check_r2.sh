#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== gpu tests ==="
timeout 420 python -m pytest tests -m gpu -x -q 2>&1 | tail -4
echo "=== solver microbench ==="
timeout 400 python scripts/solver_microbench.py 2000000
echo "=== bench 1-GPU default ==="
timeout 900 python bench.py --steps 3 --warmup 1 2> gpurun_out/bench1.log | tee gpurun_out/bench1.json
tail -2 gpurun_out/bench1.log
echo "=== bench serve mode ==="
timeout 600 python bench.py --mode serve --steps 20 --warmup 5 2> gpurun_out/bench_serve.log | tee gpurun_out/bench_serve.json
echo "=== rocprof kernel stats ==="
export TMPDIR=/tmp
cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof -o als -- \
  python /root/repo/bench.py --steps 2 --warmup 1 --users-per-gpu 2000000 --items 2000000 \
  > /root/repo/gpurun_out/bench_prof.json 2> /root/repo/gpurun_out/prof_bench.log || tail -5 /root/repo/gpurun_out/prof_bench.log
for f in /root/repo/gpurun_out/prof/*kernel_stats*; do echo "--- $f"; head -8 "$f"; done
