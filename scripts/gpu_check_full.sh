#!/bin/bash
# Comprehensive round validation: tests, smoke, benches, microbench, TCC
set -x
cd /root/repo
mkdir -p gpurun_out
echo "=== gpu tests ==="
timeout 420 python -m pytest tests -m gpu -x -q 2>&1 | tail -2
echo "=== smoke ==="
timeout 300 python -c "import __graft_entry__ as g; g.smoke()" 2>&1 | tail -2
echo "=== bench train (5 steps) ==="
timeout 600 python bench.py --steps 5 --warmup 2 2>/dev/null
echo "=== bench serve ==="
timeout 300 python bench.py --mode serve --steps 20 --warmup 5 2>/dev/null
echo "=== bench rank32 train ==="
timeout 400 python bench.py --steps 3 --warmup 1 --rank 32 --users-per-gpu 4000000 --items 4000000 2>/dev/null
echo "=== microbench ==="
timeout 300 python scripts/solver_microbench.py 2000000 2>&1 | grep -v libdrm
echo "=== TCC fetch counters (woodbury) ==="
export TMPDIR=/tmp
cd /tmp
timeout 400 rocprofv3 --pmc FETCH_SIZE WRITE_SIZE SQ_BUSY_CYCLES GRBM_GUI_ACTIVE --kernel-include-regex woodbury -d /root/repo/gpurun_out/pmc3 -o wb --output-format csv -- python -c "
import sys; sys.path.insert(0, '/root/repo')
import torch, math
from predictionio_amd.ops import als as als_ops
g = torch.Generator().manual_seed(7)
dev = torch.device('cuda')
n_rows, n_cols, f, npr = 2000000, 1000000, 64, 20
nnz = n_rows*npr
indptr = torch.arange(0, nnz+1, npr, dtype=torch.int64)[:n_rows+1].to(dev)
indices = torch.randint(0, n_cols, (nnz,), generator=g, dtype=torch.int32).to(dev)
values = torch.ones(nnz, device=dev)
Y = (torch.randn((n_cols, f), generator=g)/math.sqrt(f)).float().to(dev)
YtY = als_ops.gramian(Y)
for _ in range(2):
    X = als_ops.als_solve(indptr, indices, values, Y, YtY=YtY, lam=0.01, alpha=40.0, implicit=True)
torch.cuda.synchronize(); print('tcc workload done')
" > /root/repo/gpurun_out/pmc3.log 2>&1 || tail -3 /root/repo/gpurun_out/pmc3.log
ls /root/repo/gpurun_out/pmc3/ 2>/dev/null
