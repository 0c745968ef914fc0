"""Masked top-K scoring: fused HIP kernel on GPU, torch reference on CPU.

Replaces the reference's serve-time scoring:
recommendation recommendProductsWithFilter (ALSModel.scala:44-60),
similarproduct cosine top-N (ALSAlgorithm.scala:168-242), ecommerce
predictKnownUser/predictSimilar (ECommAlgorithm.scala:471-599).
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch
from torch.utils.weak import WeakTensorKeyDictionary

from predictionio_amd.ops.als import pad_rank

# cached bf16 copies of factor matrices for the MFMA scoring path,
# keyed weakly by the fp32 tensor (invalidated on in-place writes via
# _version). Serving keeps Y resident across batches, so the one-time
# fp32->bf16 cast amortizes to zero.
_bf16_cache: "WeakTensorKeyDictionary" = WeakTensorKeyDictionary()


def bf16_copy(t: torch.Tensor, pad_to: Optional[int] = None) -> torch.Tensor:
    """bf16 (optionally column-padded) copy of `t`, cached weakly on `t`
    and invalidated by in-place writes (_version)."""
    f = t.shape[-1]
    pad_to = pad_to or f
    ent = _bf16_cache.get(t)
    if ent is not None and ent[0] == t._version and ent[1] == pad_to:
        return ent[2]
    p = t if pad_to == f else torch.nn.functional.pad(t, (0, pad_to - f))
    b = p.to(torch.bfloat16).contiguous()
    _bf16_cache[t] = (t._version, pad_to, b)
    return b


def _mfma_rank(f: int) -> Optional[int]:
    """Padded rank for the MFMA kernel (K-dim multiples of 32), or None
    when unsupported."""
    for s in (32, 64, 128):
        if f <= s:
            return s
    return None


def topk_score(Xq: torch.Tensor, Y: torch.Tensor, K: int,
               item_mask: Optional[torch.Tensor] = None,
               ban_indptr: Optional[torch.Tensor] = None,
               ban_indices: Optional[torch.Tensor] = None,
               n_slices: Optional[int] = None,
               mode: Optional[str] = None
               ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Top-K of (Xq @ Y.T) per query row, with optional global item mask
    (uint8, 1=banned) and per-query banned lists (CSR int32, sorted).
    Returns (values [B,K] fp32, indices [B,K] int64), sorted descending.
    Banned/empty slots carry -inf / -1.

    mode: "mfma" (default on GPU) scores on the matrix cores with bf16
    inputs / fp32 accumulate, then re-checks the surviving candidates
    against the fp32 factors so returned values are exact fp32 dots;
    "fp32" forces the all-fp32 VALU kernel (v3). CPU ignores mode."""
    B, f = Xq.shape
    N = Y.shape[0]
    K = int(K)
    if Xq.is_cuda:
        if K > 64:
            # the fused kernels keep K-entry per-query lists in LDS and
            # cap at 64; arbitrary `num` (the reference allows any) falls
            # back to a materialized masked matmul + torch.topk on device
            return _topk_torch_gpu(Xq, Y, K, item_mask, ban_indptr,
                                   ban_indices)
        if mode is None:
            mode = os.environ.get("PIO_TOPK_MODE", "mfma")
        if mode == "mfma" and _mfma_rank(f) is not None:
            return _topk_score_mfma(Xq, Y, K, item_mask, ban_indptr,
                                    ban_indices, n_slices)
        from predictionio_amd.ops import hip_ext
        pf = pad_rank(f)
        Xp = Xq if pf == f else torch.nn.functional.pad(Xq, (0, pf - f))
        Yp = Y if pf == f else torch.nn.functional.pad(Y, (0, pf - f))
        if n_slices is None:
            # enough (slice, ublock) workgroups to oversubscribe 256 CUs
            # (~2048 WGs); more slices only add insert + merge cost, so
            # scale them inversely with the user-block count
            ublocks = (B + 63) // 64
            n_slices = max(1, min(2048 // ublocks + 1,
                                  (N + 255) // 256))
        vals, idxs = hip_ext().topk_score(
            Xp.contiguous(), Yp.contiguous(), K, int(n_slices),
            item_mask.contiguous() if item_mask is not None else None,
            ban_indptr.contiguous() if ban_indptr is not None else None,
            ban_indices.contiguous() if ban_indices is not None else None, 0)
        # phase 2: merge per-slice candidates (small [B, n_slices*K] topk)
        mvals, pos = torch.topk(vals, K, dim=1)
        midx = torch.gather(idxs, 1, pos).long()
        # the kernel pads empty slots with -FLT_MAX (not -inf); normalize
        # to the CPU reference's -inf / -1 convention for exact parity
        empty = mvals <= torch.finfo(torch.float32).min
        midx[empty] = -1
        mvals = mvals.masked_fill(empty, float("-inf"))
        return mvals, midx
    return topk_score_ref(Xq, Y, K, item_mask, ban_indptr, ban_indices)


def _topk_torch_gpu(Xq: torch.Tensor, Y: torch.Tensor, K: int,
                    item_mask, ban_indptr, ban_indices
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Device fallback for K > 64: materialized masked score matrix +
    torch.topk. Memory: B x N fp32 — callers with huge B should batch."""
    scores = Xq @ Y.t()
    if item_mask is not None:
        scores = scores.masked_fill(item_mask.bool().unsqueeze(0),
                                    float("-inf"))
    if ban_indptr is not None:
        ip = ban_indptr.tolist()
        for b in range(scores.shape[0]):
            banned = ban_indices[ip[b]:ip[b + 1]].long()
            if banned.numel():
                scores[b, banned] = float("-inf")
    Kc = min(K, scores.shape[1])
    vals, idxs = torch.topk(scores, Kc, dim=1)
    idxs = idxs.clone()
    idxs[vals == float("-inf")] = -1
    if Kc < K:
        B = scores.shape[0]
        vals = torch.cat(
            [vals, torch.full((B, K - Kc), float("-inf"),
                              device=vals.device)], 1)
        idxs = torch.cat(
            [idxs, torch.full((B, K - Kc), -1, dtype=idxs.dtype,
                              device=idxs.device)], 1)
    return vals, idxs


def _topk_score_mfma(Xq: torch.Tensor, Y: torch.Tensor, K: int,
                     item_mask, ban_indptr, ban_indices,
                     n_slices: Optional[int]
                     ) -> Tuple[torch.Tensor, torch.Tensor]:
    """MFMA path: bf16 matrix-core scoring selects ~2K candidates per
    query (masks/bans applied in-kernel), then the candidates are
    re-scored exactly against the fp32 factors and the final top-K taken
    on those exact values. bf16 rounding perturbs scores by ~0.4%
    relative (profiles/bf16_numerics_study.txt: 99.7% top-20 overlap
    before any margin), so with a 2x candidate margin + fp32 re-check the
    served (item, score) pairs match the fp32 reference except for
    pathological near-ties at the candidate cut."""
    from predictionio_amd.ops import hip_ext
    B, f = Xq.shape
    N = Y.shape[0]
    pf = _mfma_rank(f)
    Xp = Xq if pf == f else torch.nn.functional.pad(Xq, (0, pf - f))
    if n_slices is None:
        # just enough slices to fill the chip ONCE: the resident-WG
        # count depends on the kernel's LDS (64-item tile + the
        # per-query top-K lists), so the target shrinks for large K —
        # a 1536-WG grid at K=64 only fits 768 at a time and the
        # second pass doubled the time (profiles/serve_k_sweep_r2.log)
        upw = 128 if os.environ.get("PIO_TOPK_QB") == "2" else 64
        ublocks = (B + upw - 1) // upw
        lds = 64 * 2 * pf + 8 * (upw // 64) * 64 * (K + 1) + 4 * 64
        target = 256 * max(1, min(6, (160 * 1024) // lds))
        n_slices = max(2, min(target // ublocks, (N + 255) // 256))
    # Y's bf16 copy is cached (factors are static across serving batches);
    # Xq is cast per call — it changes every batch, and under hipGraph
    # capture (GraphedTopK) the cast must be part of the captured work
    vals, idxs = hip_ext().topk_score_mfma(
        Xp.to(torch.bfloat16).contiguous(), bf16_copy(Y, pf), K,
        int(n_slices),
        item_mask.contiguous() if item_mask is not None else None,
        ban_indptr.contiguous() if ban_indptr is not None else None,
        ban_indices.contiguous() if ban_indices is not None else None, 0)
    # merge candidate groups down to a 2K shortlist on the bf16 scores
    C = min(vals.shape[1], 2 * K)
    cv, pos = torch.topk(vals, C, dim=1)
    cidx = torch.gather(idxs, 1, pos).long()
    valid = cv > torch.finfo(torch.float32).min
    # exact fp32 re-score of the shortlist (tiny gather + batched dot)
    Yc = Y[cidx.clamp_min(0)]                      # B x C x f fp32
    exact = torch.einsum("bf,bcf->bc", Xq, Yc)
    exact = exact.masked_fill(~valid, float("-inf"))
    mv, p2 = torch.topk(exact, min(K, C), dim=1)
    midx = torch.gather(cidx, 1, p2)
    midx[mv == float("-inf")] = -1
    if mv.shape[1] < K:  # degenerate tiny-N case: pad like the reference
        pad = K - mv.shape[1]
        mv = torch.cat([mv, torch.full((B, pad), float("-inf"),
                                       device=mv.device)], 1)
        midx = torch.cat([midx, torch.full((B, pad), -1, dtype=midx.dtype,
                                           device=midx.device)], 1)
    return mv, midx


def topk_score_ref(Xq, Y, K, item_mask=None, ban_indptr=None,
                   ban_indices=None) -> Tuple[torch.Tensor, torch.Tensor]:
    """Pure-torch reference (CPU path + GPU numerics baseline)."""
    scores = Xq.float() @ Y.float().t()  # B x N
    if item_mask is not None:
        scores = scores.masked_fill(item_mask.bool().unsqueeze(0),
                                    float("-inf"))
    if ban_indptr is not None:
        ip = ban_indptr.tolist()
        for b in range(scores.shape[0]):
            banned = ban_indices[ip[b]:ip[b + 1]].long()
            if banned.numel():
                scores[b, banned] = float("-inf")
    Kc = min(K, scores.shape[1])
    vals, idxs = torch.topk(scores, Kc, dim=1)
    idxs = idxs.clone()
    idxs[vals == float("-inf")] = -1
    if Kc < K:  # pad to K like the GPU kernel (empty slots = -inf / -1)
        B = scores.shape[0]
        vals = torch.cat([vals, torch.full((B, K - Kc), float("-inf"))], 1)
        idxs = torch.cat([idxs, torch.full((B, K - Kc), -1,
                                           dtype=idxs.dtype)], 1)
    return vals, idxs


def cosine_topk(query_vec: torch.Tensor, Y_normed: torch.Tensor, K: int,
                **kw) -> Tuple[torch.Tensor, torch.Tensor]:
    """Item-item cosine kNN (K4): score(j) = sum_q cos(y_q, y_j) collapses
    to a single dot with the summed normalized query vector, so it reuses
    the top-K scoring kernel (reference computes a scalar cosine loop per
    item, similarproduct ALSAlgorithm.scala:228-242)."""
    q = query_vec.reshape(1, -1) if query_vec.dim() == 1 else query_vec
    return topk_score(q, Y_normed, K, **kw)


def normalize_rows(Y: torch.Tensor, eps: float = 1e-12) -> torch.Tensor:
    return Y / Y.norm(dim=1, keepdim=True).clamp_min(eps)
