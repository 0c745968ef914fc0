#!/usr/bin/env python3
"""Self-timed phase breakdown of the Woodbury kernel (wall_clock64 probes
on every 1024th row): stage / G-build / M-solve / emit averages, plus the
derived total-vs-counted gap (scheduling/occupancy residue)."""
import os
import sys
import math
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from predictionio_amd.ops import als as als_ops, hip_ext


def main(n_rows=2_000_000, n_cols=1_000_000, f=64, npr=20):
    dev = torch.device("cuda")
    g = torch.Generator().manual_seed(7)
    nnz = n_rows * npr
    indptr = torch.arange(0, nnz + 1, npr,
                          dtype=torch.int64)[:n_rows + 1].to(dev)
    indices = torch.randint(0, n_cols, (nnz,), generator=g,
                            dtype=torch.int32).to(dev)
    values = torch.ones(nnz, device=dev)
    Y = (torch.randn((n_cols, f), generator=g) / math.sqrt(f)).float().to(dev)
    YtY = als_ops.gramian(Y)
    Linv, V = als_ops.woodbury_lv(Y, YtY, 0.01)
    ext = hip_ext()
    prof = torch.zeros(5, dtype=torch.uint64, device=dev)
    for _ in range(2):  # warmup
        ext.als_solve(indptr, indices, values, Y, YtY, V, 0.01, 40.0,
                      True, False, 1, None, None)
    torch.cuda.synchronize()
    prof.zero_()
    t0 = time.time()
    ext.als_solve(indptr, indices, values, Y, YtY, V, 0.01, 40.0,
                  True, False, 1, None, prof)
    torch.cuda.synchronize()
    wall = time.time() - t0
    p = prof.cpu().tolist()
    n = max(p[4], 1)
    # wall_clock64 runs at 100 MHz on CDNA
    names = ["stage", "G-build", "M-solve", "emit"]
    total = 0.0
    for i, nm in enumerate(names):
        us = p[i] / n / 100.0  # ticks(100MHz) → µs
        total += us
        print(f"{nm:8s} {us:8.2f} us/row")
    print(f"{'sum':8s} {total:8.2f} us/row over {n} sampled rows")
    derived = wall / n_rows * 1e6 * (12 * 256)  # rows in flight
    print(f"wall {wall*1e3:.2f} ms for {n_rows} rows; derived per-row "
          f"latency at 12 waves/CU x 256 CU in flight: {derived:.2f} us")


if __name__ == "__main__":
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--npr", type=int, default=20)
    ap.add_argument("--rows", type=int, default=2_000_000)
    a = ap.parse_args()
    main(n_rows=a.rows, npr=a.npr)
