#!/bin/bash
set -x
cd /root/repo
echo "=== train rank sweep ==="
timeout 300 python bench.py --steps 3 --warmup 1 --rank 16 --users-per-gpu 4000000 --items 4000000 2>/dev/null
timeout 300 python bench.py --steps 3 --warmup 1 --rank 128 --users-per-gpu 4000000 --items 4000000 2>/dev/null
echo "=== train explicit mode ==="
timeout 300 python bench.py --steps 3 --warmup 1 --explicit 2>/dev/null
echo "=== train big shard (500M nnz) ==="
timeout 500 python bench.py --steps 2 --warmup 1 --users-per-gpu 25000000 2>/dev/null
echo "=== serve batch sweep ==="
for b in 1024 8192 16384; do
  timeout 200 python bench.py --mode serve --serve-batch $b --steps 10 --warmup 3 2>/dev/null
done
echo "=== pmc capture (current kernels) ==="
export TMPDIR=/tmp; cd /tmp
timeout 400 rocprofv3 --pmc SQ_WAVES SQ_INSTS_VALU SQ_INSTS_LDS SQ_LDS_BANK_CONFLICT SQ_WAIT_ANY SQ_BUSY_CYCLES --kernel-include-regex "als_|topk_score" -d /root/repo/gpurun_out/pmc4 -o cur --output-format csv -- \
  python /root/repo/bench.py --steps 2 --warmup 1 --users-per-gpu 2000000 --items 2000000 > /dev/null 2>&1
ls /root/repo/gpurun_out/pmc4/ 2>/dev/null
