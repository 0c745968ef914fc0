#!/usr/bin/env python3
"""In-kernel phase breakdown of the serve top-K kernel (wall_clock64).

Runs the config-5 serve shape (B x 10M items, rank 64, K=20, 30-item
bans) with the PROF=true kernel instantiation and prints per-phase
shares: setup (x-load + list init), item staging (HBM->LDS), score+insert,
writeback. This is the measurement NOTES.md says round-2 serve work
should start from (ATT is unavailable on this image; SQ counters
undercount wave lifetime ~10x — see NOTES.md "Counter blind spot").

wall_clock64 ticks at 100 MHz; per-workgroup sums are atomicAdd-ed into
prof[5] = {setup, stage, score, write, blocks}.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from predictionio_amd.ops import hip_ext
from predictionio_amd.ops.als import pad_rank

WCLK_MHZ = 100.0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=4096)
    ap.add_argument("--items", type=int, default=10_000_000)
    ap.add_argument("--rank", type=int, default=64)
    ap.add_argument("--topk", type=int, default=20)
    ap.add_argument("--iters", type=int, default=5)
    args = ap.parse_args()

    dev = torch.device("cuda")
    g = torch.Generator().manual_seed(3)
    B, N, f, K = args.batch, args.items, args.rank, args.topk
    Y = torch.randn((N, f), generator=g).float().to(dev)
    Xq = torch.randn((B, f), generator=g).float().to(dev)
    bans = torch.randint(0, N, (B, 30), generator=g).sort(1)[0]
    bi = torch.arange(0, 30 * (B + 1), 30, dtype=torch.int64)[:B + 1].to(dev)
    bx = bans.to(torch.int32).flatten().to(dev)
    ublocks = (B + 63) // 64
    n_slices = max(1, min(2048 // ublocks + 1, (N + 255) // 256))
    prof = torch.zeros(5, dtype=torch.uint64, device=dev)

    ext = hip_ext()
    for _ in range(2):  # warmup, unprofiled
        ext.topk_score(Xq, Y, K, n_slices, None, bi, bx, 0)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.iters):
        ext.topk_score(Xq, Y, K, n_slices, None, bi, bx, 0, prof)
    torch.cuda.synchronize()
    wall_ms = (time.time() - t0) / args.iters * 1e3

    p = prof.cpu().tolist()
    blocks = p[4]
    names = ["setup", "stage", "score", "write"]
    tot = sum(p[:4])
    print(f"B={B} N={N} f={f} K={K} n_slices={n_slices} "
          f"blocks={blocks} ({args.iters} iters)  wall {wall_ms:.2f} ms/batch")
    for i, n in enumerate(names):
        us = p[i] / WCLK_MHZ / blocks if blocks else 0.0
        print(f"  {n:6s} {us:10.2f} us/block  {100.0 * p[i] / tot:5.1f}%")
    print(f"  sum    {tot / WCLK_MHZ / blocks if blocks else 0:10.2f} us/block"
          f" (per-WG in-kernel; {blocks // args.iters} WGs/iter)")


if __name__ == "__main__":
    main()
