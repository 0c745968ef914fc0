#!/usr/bin/env python3
"""A/B the cross-slice global-threshold exchange (PIO_TOPK_GTH).

With per-slice thresholds each of a query's n_slices WGs re-pays the
full insert ramp from -inf (insert volume ~linear in n_slices, the
round-2 n_slices-sweep constraint). The global [B] atomicMax cell lets
concurrently-running slice WGs share their K-th-best through L2.

Run on a GPU box: python scripts/mfma_gth_ab.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from predictionio_amd.ops import topk as topk_ops


def check(B, N, f, K, gth, bans=False, mask=False, n_slices=None):
    g = torch.Generator().manual_seed(B * 31 + N * 7 + K)
    dev = torch.device("cuda")
    Xq = torch.randn((B, f), generator=g).float().to(dev)
    Y = torch.randn((N, f), generator=g).float().to(dev)
    bi = bx = im = None
    if bans:
        nb = torch.randint(0, 8, (B,), generator=g)
        bi = torch.zeros(B + 1, dtype=torch.int64)
        bi[1:] = nb.cumsum(0)
        bxl = torch.randint(0, N, (int(bi[-1]),), generator=g,
                            dtype=torch.int32)
        rows = [torch.sort(bxl[bi[b]:bi[b + 1]])[0] for b in range(B)]
        bx = (torch.cat(rows) if rows else bxl).to(dev)
        bi = bi.to(dev)
    if mask:
        im = (torch.rand(N, generator=g) < 0.1).to(torch.uint8).to(dev)
    os.environ["PIO_TOPK_GTH"] = "1" if gth else "0"
    mv, mi = topk_ops.topk_score(Xq, Y, K, item_mask=im, ban_indptr=bi,
                                 ban_indices=bx, mode="mfma",
                                 n_slices=n_slices)
    fv, fi = topk_ops.topk_score(Xq, Y, K, item_mask=im, ban_indptr=bi,
                                 ban_indices=bx, mode="fp32")
    bad = 0
    for b in range(B):
        if set(mi[b].tolist()) != set(fi[b].tolist()):
            if not torch.allclose(mv[b], fv[b], atol=1e-3, rtol=1e-4):
                bad += 1
    tag = (f"B={B} N={N} f={f} K={K} gth={gth} bans={bans} mask={mask} "
           f"ns={n_slices}")
    print(f"check {tag}: {'OK' if bad == 0 else f'{bad} BAD LISTS'}")
    return bad


def bench(B, N, f, K, gth, n_slices=None, iters=10, warmup=3, bans=False):
    g = torch.Generator().manual_seed(7)
    dev = torch.device("cuda")
    Xq = torch.randn((B, f), generator=g).float().to(dev)
    Y = torch.randn((N, f), generator=g).float().to(dev)
    bi = bx = None
    if bans:
        bi = (torch.arange(B + 1, dtype=torch.int64) * 8).to(dev)
        bxl = torch.randint(0, N, (B * 8,), generator=g, dtype=torch.int32)
        bx = torch.sort(bxl.view(B, 8), dim=1)[0].reshape(-1).to(dev)
    os.environ["PIO_TOPK_GTH"] = "1" if gth else "0"
    for _ in range(warmup):
        topk_ops.topk_score(Xq, Y, K, mode="mfma", n_slices=n_slices,
                            ban_indptr=bi, ban_indices=bx)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        topk_ops.topk_score(Xq, Y, K, mode="mfma", n_slices=n_slices,
                            ban_indptr=bi, ban_indices=bx)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / iters
    print(f"bench B={B} gth={int(gth)} ns={n_slices} bans={int(bans)}: "
          f"{dt * 1e3:8.2f} ms  {B / dt:10.0f} q/s")
    return dt


def main():
    bad = 0
    for gth in (False, True):
        bad += check(100, 200_000, 64, 20, gth)
        bad += check(129, 100_000, 64, 20, gth, bans=True, mask=True)
        bad += check(256, 50_000, 32, 1, gth)
        bad += check(2048, 500_000, 64, 20, gth, mask=True)
        bad += check(512, 1_000_000, 64, 20, gth, n_slices=64)
    if bad:
        print(f"TOTAL BAD: {bad}")
        sys.exit(1)
    print("--- timing (10M items, f=64, K=20) ---")
    N, f, K = 10_000_000, 64, 20
    for B in (4096, 16384):
        for bans in (False, True):
            bench(B, N, f, K, gth=False, bans=bans)
            for ns in (None, 48, 96):
                bench(B, N, f, K, gth=True, n_slices=ns, bans=bans)
    for B in (1, 256, 1024):
        bench(B, N, f, K, gth=False)
        bench(B, N, f, K, gth=True)
    os.environ.pop("PIO_TOPK_GTH", None)


if __name__ == "__main__":
    main()
