#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== gpu tests ==="
timeout 420 python -m pytest tests -m gpu -x -q 2>&1 | tail -3
echo "=== solver microbench ==="
timeout 300 python scripts/solver_microbench.py 2000000 2>&1 | grep -v libdrm | head -4
echo "=== bench train ==="
timeout 600 python bench.py --steps 3 --warmup 1 2>gpurun_out/b1.log | tee gpurun_out/bench1.json
echo "=== rocprof kernel stats (train small) ==="
export TMPDIR=/tmp
cd /tmp
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof2 -o als -- \
  python /root/repo/bench.py --steps 2 --warmup 1 --users-per-gpu 2000000 --items 2000000 \
  > /dev/null 2>&1
head -4 /root/repo/gpurun_out/prof2/als_kernel_stats.csv | cut -c1-120
echo "=== pmc counters (woodbury + topk) ==="
timeout 400 rocprofv3 --pmc SQ_WAVES SQ_INSTS_VALU SQ_INSTS_LDS SQ_LDS_BANK_CONFLICT SQ_WAIT_ANY SQ_BUSY_CYCLES -d /root/repo/gpurun_out/pmc -o als --output-format csv -- \
  python -c "
import sys; sys.path.insert(0, '/root/repo')
import torch, time
from predictionio_amd.ops import als as als_ops, topk as topk_ops
import math
g = torch.Generator().manual_seed(7)
dev = torch.device('cuda')
n_rows, n_cols, f, npr = 1000000, 1000000, 64, 20
nnz = n_rows*npr
indptr = torch.arange(0, nnz+1, npr, dtype=torch.int64)[:n_rows+1].to(dev)
indices = torch.randint(0, n_cols, (nnz,), generator=g, dtype=torch.int32).to(dev)
values = torch.ones(nnz, device=dev)
Y = (torch.randn((n_cols, f), generator=g)/math.sqrt(f)).float().to(dev)
YtY = als_ops.gramian(Y)
for _ in range(3):
    X = als_ops.als_solve(indptr, indices, values, Y, YtY=YtY, lam=0.01, alpha=40.0, implicit=True)
Xq = torch.randn((4096, f), generator=g).float().to(dev)
for _ in range(3):
    v, i = topk_ops.topk_score(Xq, Y, 20)
torch.cuda.synchronize()
print('pmc workload done')
" > /root/repo/gpurun_out/pmc.log 2>&1 || tail -5 /root/repo/gpurun_out/pmc.log
ls /root/repo/gpurun_out/pmc/ 2>/dev/null
