#!/usr/bin/env python3
"""A/B the serve top-K paths at the config-5 shape: v3 fp32 VALU kernel
vs v5 MFMA (bf16 score + fp32 rescore of survivors).

Prints ms/batch + q/s for each mode over a batch sweep, plus a recall
check of mfma vs fp32 on a subsample. Run on a GPU box:
  python scripts/serve_mfma_ab.py --items 10000000 --batches 4096
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from predictionio_amd.ops import topk as topk_ops


def bench_mode(Xq, Y, K, bi, bx, mode, iters=5, warmup=2):
    for _ in range(warmup):
        topk_ops.topk_score(Xq, Y, K, ban_indptr=bi, ban_indices=bx,
                            mode=mode)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        v, i = topk_ops.topk_score(Xq, Y, K, ban_indptr=bi, ban_indices=bx,
                                   mode=mode)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / iters
    return dt, v, i


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--items", type=int, default=10_000_000)
    ap.add_argument("--rank", type=int, default=64)
    ap.add_argument("--topk", type=int, default=20)
    ap.add_argument("--batches", type=int, nargs="+",
                    default=[1, 256, 1024, 4096, 16384])
    ap.add_argument("--iters", type=int, default=5)
    ap.add_argument("--check", action="store_true",
                    help="recall check vs exact torch on a small shape")
    args = ap.parse_args()

    dev = torch.device("cuda")
    g = torch.Generator().manual_seed(7)
    Y = torch.randn((args.items, args.rank), generator=g).float().to(dev)
    K = args.topk

    if args.check:
        Ns, Bs = 100_000, 256
        Ys = Y[:Ns]
        Xs = torch.randn((Bs, args.rank), generator=g).float().to(dev)
        fv, fi = topk_ops.topk_score(Xs, Ys, K, mode="fp32")
        mv, mi = topk_ops.topk_score(Xs, Ys, K, mode="mfma")
        hits = sum(len(set(mi[b].tolist()) & set(fi[b].tolist()))
                   for b in range(Bs))
        print(f"recall mfma-vs-fp32 @K={K}, N={Ns}: {hits / (Bs * K):.4f}")
        ok = torch.allclose(mv, fv, atol=1e-3, rtol=1e-3)
        print(f"value agreement (1e-3): {ok}")

    for B in args.batches:
        Xq = torch.randn((B, args.rank), generator=g).float().to(dev)
        bans = torch.randint(0, args.items, (B, 30), generator=g) \
            .sort(1)[0].to(torch.int32)
        bi = torch.arange(0, 30 * (B + 1), 30, dtype=torch.int64)[:B + 1] \
            .to(dev)
        bx = bans.flatten().to(dev)
        out = {}
        for mode in ("fp32", "mfma"):
            dt, v, i = bench_mode(Xq, Y, K, bi, bx, mode,
                                  iters=args.iters)
            out[mode] = dt
            print(f"B={B:6d} mode={mode}: {dt * 1e3:8.2f} ms/batch  "
                  f"{B / dt:10.0f} q/s", flush=True)
        print(f"B={B:6d} speedup mfma/fp32: {out['fp32'] / out['mfma']:.2f}x",
              flush=True)


if __name__ == "__main__":
    main()
