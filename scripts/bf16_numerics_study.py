#!/usr/bin/env python3
"""Numerics study for the two round-2 bf16 levers (runs on CPU).

NOTES.md lists two bandwidth/occupancy levers that both hinge on the
same open question — does bf16 in the factor path hurt ALS quality?
  (a) bf16 LDS staging of V/Y inside the solver kernels (halves the
      13.3 KB/wave Yl buffer -> 2x resident waves for the Woodbury);
  (b) bf16 RCCL all-gather of the factor shards (halves the 25.6 GB
      item-half-step gather at 8 GPUs).

This script simulates both with the torch fp32 reference solver on a
small implicit-ALS problem and reports, per variant:
  - weighted implicit objective per iteration (Hu-Koren loss), and
  - top-20 recommendation overlap vs the fp32 run after training.
Casting POLICY matters: in both levers the Gramian/YtY accumulation and
the n x n solve stay fp32 — only the staged/communicated factor VALUES
are rounded. That is what's simulated (round-trip through bf16 at the
half-step boundary / before the per-row dot products).
"""
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from predictionio_amd.ops import als as als_ops


def implicit_loss(X, Y, indptr, indices, values, lam, alpha):
    """Hu-Koren weighted objective, full-matrix form (small problems):
    sum_ui c_ui (p_ui - x_u.y_i)^2 + lam (|X|^2 + |Y|^2), with c=1+alpha*r
    on observed cells, c=1 / p=0 elsewhere."""
    S = X @ Y.t()
    loss = (S ** 2).sum()  # unobserved: c=1, p=0
    for u in range(indptr.numel() - 1):
        for j in range(int(indptr[u]), int(indptr[u + 1])):
            i = int(indices[j])
            r = float(values[j])
            s = float(S[u, i])
            loss += (1 + alpha * r) * (1 - s) ** 2 - s * s
    return float(loss + lam * ((X ** 2).sum() + (Y ** 2).sum()))


def run(n_users=600, n_items=400, nnz_per=12, f=32, iters=8, lam=0.05,
        alpha=4.0, mode="fp32", seed=7):
    g = torch.Generator().manual_seed(seed)
    users = torch.arange(n_users, dtype=torch.int32).repeat_interleave(nnz_per)
    items = torch.randint(0, n_items, (n_users * nnz_per,), generator=g,
                          dtype=torch.int32)
    vals = torch.ones(n_users * nnz_per)
    u, i, v = als_ops.aggregate_ratings(users, items, vals, n_items, "sum")
    up, ui, uv = als_ops.build_csr(u, i, v, n_users)
    ip, iu, iv = als_ops.build_csr(i, u, v, n_items)
    X = torch.randn((n_users, f), generator=g) / math.sqrt(f)
    Y = torch.randn((n_items, f), generator=g) / math.sqrt(f)

    def stage(t):
        # the bf16 lever: factor values rounded at the staging boundary
        return t.to(torch.bfloat16).float() if mode != "fp32" else t

    losses = []
    for _ in range(iters):
        Ys = stage(Y)
        X = als_ops.als_solve_ref(up, ui, uv, Ys, YtY=als_ops.gramian(Ys),
                                  lam=lam, alpha=alpha, implicit=True)
        Xs = stage(X)
        Y = als_ops.als_solve_ref(ip, iu, iv, Xs, YtY=als_ops.gramian(Xs),
                                  lam=lam, alpha=alpha, implicit=True)
        losses.append(implicit_loss(X, Y, up, ui, uv, lam, alpha))
    return X, Y, losses


def topk_overlap(Xa, Ya, Xb, Yb, K=20):
    ia = torch.topk(Xa @ Ya.t(), K, dim=1)[1]
    ib = torch.topk(Xb @ Yb.t(), K, dim=1)[1]
    inter = torch.tensor([
        len(set(ia[r].tolist()) & set(ib[r].tolist()))
        for r in range(ia.shape[0])], dtype=torch.float32)
    return float(inter.mean()) / K


if __name__ == "__main__":
    X0, Y0, l0 = run(mode="fp32")
    X1, Y1, l1 = run(mode="bf16")
    print("iter  fp32-loss      bf16-staged-loss   rel-diff")
    for k, (a, b) in enumerate(zip(l0, l1)):
        print(f"{k:4d}  {a:13.2f}  {b:17.2f}  {abs(a - b) / abs(a):9.2e}")
    ov = topk_overlap(X0, Y0, X1, Y1)
    print(f"top-20 overlap fp32 vs bf16-staged: {ov * 100:.2f}%")
