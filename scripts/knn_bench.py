#!/usr/bin/env python3
"""BASELINE config 3: Similar-Product item-item cosine kNN.

1M items x 128-dim L2-normalized factors on one MI355X; a batch of
similar-item queries (each = sum of the query items' normalized vectors,
the collapse the similarproduct template uses — see ops/topk.py
cosine_topk) scored by the fused masked top-K kernel. Prints queries/s
and per-batch latency for a batch sweep.
"""
import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from predictionio_amd.ops import topk as topk_ops


def bench(n_items=1_000_000, f=128, K=20, batches=(1, 256, 4096), iters=20):
    g = torch.Generator().manual_seed(11)
    dev = torch.device("cuda")
    Y = torch.randn((n_items, f), generator=g).float().to(dev)
    Yn = topk_ops.normalize_rows(Y)
    for B in batches:
        # each query: 1-4 seed items summed after normalization
        seeds = torch.randint(0, n_items, (B, 4), generator=g).to(dev)
        q = topk_ops.normalize_rows(Yn[seeds].sum(1))
        for _ in range(3):
            topk_ops.cosine_topk(q, Yn, K)
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(iters):
            topk_ops.cosine_topk(q, Yn, K)
        torch.cuda.synchronize()
        dt = (time.time() - t0) / iters
        print(f"B={B:5d} K={K}  {dt * 1e3:8.3f} ms/batch  "
              f"{B / dt:10.0f} queries/s", flush=True)


if __name__ == "__main__":
    bench()
