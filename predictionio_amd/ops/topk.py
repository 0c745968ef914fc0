"""Masked top-K scoring: fused HIP kernel on GPU, torch reference on CPU.

Replaces the reference's serve-time scoring:
recommendation recommendProductsWithFilter (ALSModel.scala:44-60),
similarproduct cosine top-N (ALSAlgorithm.scala:168-242), ecommerce
predictKnownUser/predictSimilar (ECommAlgorithm.scala:471-599).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from predictionio_amd.ops.als import pad_rank


def topk_score(Xq: torch.Tensor, Y: torch.Tensor, K: int,
               item_mask: Optional[torch.Tensor] = None,
               ban_indptr: Optional[torch.Tensor] = None,
               ban_indices: Optional[torch.Tensor] = None,
               n_slices: Optional[int] = None
               ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Top-K of (Xq @ Y.T) per query row, with optional global item mask
    (uint8, 1=banned) and per-query banned lists (CSR int32, sorted).
    Returns (values [B,K] fp32, indices [B,K] int64), sorted descending.
    Banned/empty slots carry -inf / -1."""
    B, f = Xq.shape
    N = Y.shape[0]
    K = int(K)
    if Xq.is_cuda:
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
