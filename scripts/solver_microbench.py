#!/usr/bin/env python3
"""Time the fused ALS Gramian+Cholesky solver kernel in isolation.

Compares iteration cost across nnz-per-row to show whether the per-row
fixed (solve) cost or the streaming (Gramian) cost dominates, and prints
achieved rows/s + effective HBM GB/s for the Gramian read stream.
"""
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from predictionio_amd.ops import als as als_ops


def bench(n_rows, n_cols, f, nnz_per_row, implicit=True, iters=10):
    g = torch.Generator().manual_seed(7)
    device = torch.device("cuda")
    nnz = n_rows * nnz_per_row
    indptr = torch.arange(0, nnz + 1, nnz_per_row, dtype=torch.int64)[: n_rows + 1].to(device)
    indices = torch.randint(0, n_cols, (nnz,), generator=g, dtype=torch.int32).to(device)
    values = torch.ones(nnz, device=device)
    Y = (torch.randn((n_cols, f), generator=g) / math.sqrt(f)).float().to(device)
    YtY = als_ops.gramian(Y) if implicit else None
    for _ in range(3):
        X = als_ops.als_solve(indptr, indices, values, Y, lam=0.01, alpha=40.0,
                              implicit=implicit, YtY=YtY)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        X = als_ops.als_solve(indptr, indices, values, Y, lam=0.01, alpha=40.0,
                              implicit=implicit, YtY=YtY)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / iters
    rows_s = n_rows / dt
    gbs = nnz * f * 4 / dt / 1e9  # factor-row reads only (min traffic)
    print(f"f={f:3d} rows={n_rows/1e6:5.1f}M nnz/row={nnz_per_row:3d} "
          f"{dt*1e3:8.2f} ms  {rows_s/1e6:7.2f} Mrows/s  {gbs:7.1f} GB/s(Y-reads)",
          flush=True)
    return dt


if __name__ == "__main__":
    n_rows = int(sys.argv[1]) if len(sys.argv) > 1 else 2_000_000
    n_cols = 1_000_000
    for f in (64, 128):
        for nnz_per_row in (5, 20, 40):
            bench(n_rows, n_cols, f, nnz_per_row)
    # the 8-GPU item-side shape: rows densify with world size
    for nnz_per_row in (100, 200):
        bench(n_rows // 4, n_cols, 64, nnz_per_row)
