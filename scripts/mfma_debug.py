#!/usr/bin/env python3
"""Debug: raw topk_score_mfma kernel candidates vs per-slice brute force
on bf16-rounded inputs. Pinpoints which (query, slice) lists are wrong."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from predictionio_amd.ops import hip_ext


def check(B, N, f, K, ns, seed):
    g = torch.Generator().manual_seed(seed)
    Xq = torch.randn((B, f), generator=g).float()
    Y = torch.randn((N, f), generator=g).float()
    Xb = Xq.to(torch.bfloat16)
    Yb = Y.to(torch.bfloat16)
    # exact fp32-accum scores of bf16 inputs (matmul of bf16 upcast)
    S = Xb.float() @ Yb.float().t()  # B x N
    per = (N + ns - 1) // ns
    ext = hip_ext()
    vals, idxs = ext.topk_score_mfma(Xb.cuda().contiguous(),
                                     Yb.cuda().contiguous(),
                                     K, ns, None, None, None, 0)
    vals = vals.cpu().view(B, ns, K)
    idxs = idxs.cpu().view(B, ns, K)
    bad = 0
    for s in range(ns):
        lo, hi = s * per, min(N, s * per + per)
        if lo >= hi:
            continue
        ref_v, ref_i = S[:, lo:hi].topk(min(K, hi - lo), dim=1)
        for b in range(B):
            got = sorted(vals[b, s].tolist(), reverse=True)[:min(K, hi - lo)]
            want = ref_v[b].tolist()
            for gv, wv in zip(got, want):
                if abs(gv - wv) > 1e-3 + 1e-3 * abs(wv):
                    if bad < 10:
                        print(f"  q={b} slice={s} got={got} want={want} "
                              f"idx={idxs[b, s].tolist()}")
                    bad += 1
                    break
    print(f"B={B} N={N} f={f} K={K} ns={ns} seed={seed}: "
          f"{bad} bad (of {B * ns}) lists")
    return bad


if __name__ == "__main__":
    total = 0
    total += check(37, 5000, 32, 1, 7, 132)   # the failing test shape
    total += check(37, 5000, 32, 4, 7, 432)
    total += check(37, 5000, 64, 1, 7, 164)
    total += check(64, 5000, 64, 20, 7, 2064)
    total += check(128, 20000, 128, 20, 11, 20128)
    print("TOTAL BAD:", total)
