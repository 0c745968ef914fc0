#!/usr/bin/env python3
"""Phase breakdown (wall_clock64 PROF variant) + n_slices sweep for the
MFMA serve kernel at the config-5 shape.

Buckets: setup (x-frag load + list init), stage (barrier + LDS drain),
score (MFMA + epilogue/insert), write(back). Ticks at 100 MHz, summed
per workgroup by lane 0 of wave 0 only — shares are indicative, the
sweep timings are the ground truth.
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from predictionio_amd.ops import hip_ext, topk as topk_ops

WCLK_MHZ = 100.0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--items", type=int, default=10_000_000)
    ap.add_argument("--rank", type=int, default=64)
    ap.add_argument("--batch", type=int, default=4096)
    ap.add_argument("--topk", type=int, default=20)
    ap.add_argument("--slices", type=int, nargs="+",
                    default=[8, 16, 33, 64, 128, 256])
    args = ap.parse_args()
    dev = torch.device("cuda")
    g = torch.Generator().manual_seed(7)
    Y = torch.randn((args.items, args.rank), generator=g).float().to(dev)
    Xq = torch.randn((args.batch, args.rank), generator=g).float().to(dev)
    Yb = topk_ops.bf16_copy(Y)
    Xb = Xq.to(torch.bfloat16).contiguous()
    ext = hip_ext()
    K = args.topk

    # --- n_slices sweep (kernel only, no merge)
    for ns in args.slices:
        for _ in range(2):
            ext.topk_score_mfma(Xb, Yb, K, ns, None, None, None, 0)
        torch.cuda.synchronize()
        t0 = time.time()
        for _ in range(3):
            ext.topk_score_mfma(Xb, Yb, K, ns, None, None, None, 0)
        torch.cuda.synchronize()
        dt = (time.time() - t0) / 3
        print(f"n_slices={ns:4d}: kernel {dt * 1e3:8.2f} ms", flush=True)

    # --- phase probe at the default slicing
    ns = 33
    prof = torch.zeros(5, dtype=torch.uint64, device=dev)
    ext.topk_score_mfma(Xb, Yb, K, ns, None, None, None, 0, prof)
    torch.cuda.synchronize()
    p = prof.cpu().tolist()
    blocks = p[4]
    names = ["setup", "stage", "score", "write"]
    tot = sum(p[:4])
    print(f"\nphase probe (n_slices={ns}, {blocks} WGs sampled):")
    for i, n in enumerate(names):
        us = p[i] / WCLK_MHZ / blocks
        print(f"  {n:6s}: {us:10.1f} us/WG  ({p[i] / tot * 100:5.1f}%)")
    print(f"  sum: {tot / WCLK_MHZ / blocks:.1f} us/WG")


if __name__ == "__main__":
    main()
