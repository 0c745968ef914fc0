"""ALS half-iteration solve: HIP kernel on GPU, torch reference on CPU.

Replaces MLlib's normal-equation construction + CholeskySolver (reference:
ALS.train/trainImplicit call sites, SURVEY.md §2.7/2.9 K1+K2). The CPU path
is the numerics reference the GPU kernel is validated against.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

SUPPORTED_RANKS = (16, 32, 64, 128)


def pad_rank(f: int) -> int:
    for s in SUPPORTED_RANKS:
        if f <= s:
            return s
    raise ValueError(f"rank {f} > 128 not supported")


def gramian(Y: torch.Tensor) -> torch.Tensor:
    """YtY (f x f) — a plain library GEMM (hipBLASLt on device)."""
    return Y.t() @ Y


def woodbury_lv(Y: torch.Tensor, YtY: torch.Tensor,
                lam: float) -> tuple:
    """(Linv, V): L = chol(YtY + lam I), Linv = L^-1, V = Y L^-T — per-half-iteration
    precompute that enables the per-row Woodbury fast path
    (als_woodbury_kernel). One fxf Cholesky + one triangular-solve GEMM
    over all items (rocBLAS trsm); the whitened rows satisfy
    v_i . v_j = y_i^T B^-1 y_j."""
    f = Y.shape[1]
    eye = torch.eye(f, dtype=Y.dtype, device=Y.device)
    B = YtY + lam * eye
    L = torch.linalg.cholesky(B)
    # Materialize L^-1 (f x f, cheap) and whiten with a GEMM: hipBLAS trsm
    # with a 10M-row rhs hit HIPBLAS_STATUS_ALLOC_FAILED at bench scale,
    # while hipBLASLt GEMMs of that shape are proven.
    Linv = torch.linalg.solve_triangular(L, eye, upper=False)
    V = (Y @ Linv.mT).contiguous()  # row v_i = L^-1 y_i
    import os as _os
    if (_os.environ.get("PIO_ALS_STAGE_BF16") == "1" and Y.is_cuda):
        V = V.to(torch.bfloat16).contiguous()
    return Linv, V


def als_solve(indptr: torch.Tensor, indices: torch.Tensor,
              values: torch.Tensor, Y: torch.Tensor,
              YtY: Optional[torch.Tensor] = None,
              lam: float = 0.01, alpha: float = 1.0,
              implicit: bool = False, wr_scale: bool = True,
              lv: Optional[tuple] = None,
              row_range: Optional[Tuple[int, int]] = None,
              out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Solve all rows of one ALS half-iteration.

    explicit: (sum y y^T + lam*nnz*I) x = sum r*y      (ALS-WR, like MLlib)
    implicit: (YtY + sum alpha*r y y^T + lam*I) x = sum (1+alpha*r) y
              (Hu-Koren; YtY required)

    On GPU, rows with nnz <= 32 take the Woodbury path (exact — the same
    linear system through the push-through identity). Implicit mode runs
    three phases: the Woodbury kernel emits whitened Z rows, one rocBLAS
    triangular solve maps Z -> X for all rows, then the dense kernel fills
    the big (nnz > 32) rows directly.

    lv: optional precomputed (Linv, V) from prepare_lv() — hoist it when
    solving in row chunks (the multi-GPU gather/solve overlap path).
    row_range: solve only rows [lo, hi) of the CSR (indices/values stay
    whole; indptr offsets are absolute). out: optional [rows, f] target
    written in place (enables chunked solves into one padded buffer).
    """
    if implicit:
        wr_scale = False  # Hu-Koren regularizes with plain lambda*I
        if YtY is None:
            YtY = gramian(Y)
    if row_range is not None:
        lo, hi = row_range
        indptr = indptr[lo:hi + 1]
    if Y.is_cuda:
        from predictionio_amd.ops import hip_ext
        ext = hip_ext()
        f = Y.shape[1]
        pf = pad_rank(f)
        Yp = (Y if pf == f
              else torch.nn.functional.pad(Y, (0, pf - f))).contiguous()
        YtYp = None
        if YtY is not None:
            YtYp = (YtY if pf == f
                    else torch.nn.functional.pad(YtY, (0, pf - f, 0, pf - f))
                    ).contiguous()
        ip, ix, vv = (indptr.contiguous(), indices.contiguous(),
                      values.contiguous())
        # out-of-range column ids would make the kernels gather OOB from
        # Y (undefined behavior that can wedge the device) — fail loudly
        # here instead; one .max() reduction is noise vs the solve
        if ix.numel():
            mx = int(ix.max())
            if mx >= Y.shape[0] or int(ix.min()) < 0:
                raise ValueError(
                    f"CSR column id out of range: [{int(ix.min())}, {mx}]"
                    f" vs {Y.shape[0]} factor rows")
        direct_out = (out if out is not None and pf == f
                      and out.is_contiguous() else None)
        if implicit and pf <= 128:
            if lv is not None:
                Linv, V = lv
            else:
                Linv, V = woodbury_lv(Yp, YtYp, lam)
            # opt-in bf16 staging of the whitened factors (see the
            # BF16S kernel variant; numerics bounded by
            # profiles/bf16_numerics_study.txt)
            import os as _os
            if _os.environ.get("PIO_ALS_STAGE_BF16") == "1" \
                    and V.dtype == torch.float32:
                V = V.to(torch.bfloat16).contiguous()
            Z = ext.als_solve(ip, ix, vv, Yp, YtYp, V, float(lam),
                              float(alpha), True, False, 1, None)
            # X = Z L^-1 for all rows (Woodbury rows hold z; big rows get
            # overwritten by the dense pass next)
            if direct_out is not None:
                X = direct_out
                torch.matmul(Z, Linv, out=X)
            else:
                X = (Z @ Linv).contiguous()
            ext.als_solve(ip, ix, vv, Yp, YtYp, None, float(lam),
                          float(alpha), True, False, 2, X)
        else:
            X = ext.als_solve(ip, ix, vv, Yp, YtYp, None, float(lam),
                              float(alpha), bool(implicit), bool(wr_scale),
                              0, direct_out)
        res = X[:, :f].contiguous() if pf != f else X
        if out is not None and res.data_ptr() != out.data_ptr():
            out.copy_(res)
            return out
        return res
    res = als_solve_ref(indptr, indices, values, Y, YtY, lam, alpha,
                        implicit, wr_scale)
    if out is not None:
        out.copy_(res)
        return out
    return res


def prepare_lv(Y: torch.Tensor, YtY: torch.Tensor, lam: float):
    """Precompute the (Linv, V) Woodbury pair once per half-iteration so
    chunked solves (gather/solve overlap) don't redo the whitening GEMM
    per chunk. Pads to the supported rank like als_solve."""
    f = Y.shape[1]
    pf = pad_rank(f)
    Yp = (Y if pf == f
          else torch.nn.functional.pad(Y, (0, pf - f))).contiguous()
    YtYp = (YtY if pf == f
            else torch.nn.functional.pad(YtY, (0, pf - f, 0, pf - f))
            ).contiguous()
    return woodbury_lv(Yp, YtYp, lam)


def als_solve_ref(indptr, indices, values, Y, YtY=None, lam=0.01, alpha=1.0,
                  implicit=False, wr_scale=True) -> torch.Tensor:
    """Pure-torch fp32 reference (used on CPU and in GPU numerics tests)."""
    n_rows = indptr.shape[0] - 1
    f = Y.shape[1]
    X = torch.zeros((n_rows, f), dtype=Y.dtype, device=Y.device)
    eye = torch.eye(f, dtype=Y.dtype, device=Y.device)
    ip = indptr.tolist()
    for r in range(n_rows):
        s, e = ip[r], ip[r + 1]
        nnz = e - s
        if implicit:
            A = (YtY if YtY is not None else gramian(Y)).clone()
            b = torch.zeros(f, dtype=Y.dtype, device=Y.device)
            if nnz:
                cols = indices[s:e].long()
                Yr = Y[cols]
                v = values[s:e]
                A = A + (Yr.t() * (alpha * v)) @ Yr
                b = Yr.t() @ (1.0 + alpha * v)
            A = A + lam * eye
        else:
            if nnz == 0:
                continue
            cols = indices[s:e].long()
            Yr = Y[cols]
            v = values[s:e]
            A = Yr.t() @ Yr + lam * (nnz if wr_scale else 1.0) * eye
            b = Yr.t() @ v
        X[r] = torch.linalg.solve(A, b)
    return X


def build_csr(rows: torch.Tensor, cols: torch.Tensor, vals: torch.Tensor,
              n_rows: int) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """(row, col, val) triples → CSR (indptr i64, indices i32, values f32).

    Device-side sort-based build — torch.sort on GPU.
    """
    order = torch.argsort(rows)
    r = rows[order]
    indices = cols[order].to(torch.int32)
    values = vals[order].to(torch.float32)
    counts = torch.bincount(r.long(), minlength=n_rows)
    indptr = torch.zeros(n_rows + 1, dtype=torch.int64, device=rows.device)
    torch.cumsum(counts, 0, out=indptr[1:])
    return indptr, indices, values


def aggregate_ratings(rows: torch.Tensor, cols: torch.Tensor,
                      vals: torch.Tensor, n_cols: int,
                      mode: str = "sum"):
    """Dedup (row, col) pairs: 'sum' (e-commerce reduceByKey semantics,
    ECommAlgorithm.scala:168-205) or 'latest' (similarproduct latest-wins,
    ALSAlgorithm.scala:88-120 — caller passes vals in time order)."""
    key = rows.long() * n_cols + cols.long()
    if mode == "sum":
        uk, inv = torch.unique(key, return_inverse=True)
        out = torch.zeros(uk.numel(), dtype=torch.float32, device=vals.device)
        out.scatter_add_(0, inv, vals.float())
    elif mode == "latest":
        # stable sort keeps time order within key; take last occurrence
        sk, order = torch.sort(key, stable=True)
        last = torch.ones(sk.numel(), dtype=torch.bool, device=key.device)
        last[:-1] = sk[1:] != sk[:-1]
        uk = sk[last]
        out = vals[order][last].float()
    else:
        raise ValueError(mode)
    return (uk // n_cols).to(rows.dtype), (uk % n_cols).to(cols.dtype), out
