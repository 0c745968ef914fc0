#!/bin/bash
# Round-2 starter: advanced thread trace of the Woodbury kernel (SQ
# counters cannot localize its remaining per-row latency — NOTES.md).
# TESTED round 1: FAILS on this image — rocprofv3 aborts with
# "rocprof-trace-decoder library path not found" (decoder not shipped in
# ROCm 7.2.0 here). Round-2 fallback: instrument the kernel itself with
# s_memtime deltas written to a debug buffer per phase (stage/G/solve),
# which needs no tooling. Keep --pmc OUT of any trace invocation.
set -x
export TMPDIR=/tmp
cd /tmp
timeout 500 rocprofv3 --advanced-thread-trace --att-target-cu 0 \
  --kernel-include-regex woodbury \
  -d /root/repo/gpurun_out/att -o wb -- \
  python -c "
import sys; sys.path.insert(0, '/root/repo')
import torch, math
from predictionio_amd.ops import als as als_ops
g = torch.Generator().manual_seed(7)
dev = torch.device('cuda')
n_rows, n_cols, f, npr = 200000, 200000, 64, 20
nnz = n_rows*npr
indptr = torch.arange(0, nnz+1, npr, dtype=torch.int64)[:n_rows+1].to(dev)
indices = torch.randint(0, n_cols, (nnz,), generator=g, dtype=torch.int32).to(dev)
values = torch.ones(nnz, device=dev)
Y = (torch.randn((n_cols, f), generator=g)/math.sqrt(f)).float().to(dev)
X = als_ops.als_solve(indptr, indices, values, Y, lam=0.01, alpha=40.0, implicit=True)
torch.cuda.synchronize(); print('att workload done')
"
ls -la /root/repo/gpurun_out/att/ 2>/dev/null | head
