"""Multi-GPU sharded top-K serving (SURVEY.md §2.8 serving concurrency /
batch-predict parallelism).

Item factors are sharded by row block across the process group (one rank
per GPU); each rank scores its shard with the fused masked top-K kernel
and the per-shard candidates are merged with one all-gather + a small
torch.topk — the K-merge the reference performs on the driver after
Spark's `top(num)` per partition.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from predictionio_amd.ops import topk as topk_ops
from predictionio_amd.parallel import dist as pdist


def sharded_topk_score(Xq: torch.Tensor, Y_local: torch.Tensor, K: int,
                       item_base: int,
                       item_mask: Optional[torch.Tensor] = None,
                       ban_indptr: Optional[torch.Tensor] = None,
                       ban_indices: Optional[torch.Tensor] = None
                       ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Top-K over the UNION of all ranks' item shards.

    Xq is replicated (every rank scores the same query batch); Y_local is
    this rank's item rows, whose global ids start at `item_base`. Ban
    lists are in LOCAL shard coordinates. Returns (values, global indices)
    identical on every rank.
    """
    v, idx = topk_ops.topk_score(Xq, Y_local, K, item_mask=item_mask,
                                 ban_indptr=ban_indptr,
                                 ban_indices=ban_indices)
    gidx = torch.where(idx >= 0, idx + item_base, idx)
    if not pdist.is_distributed():
        return v, gidx
    import torch.distributed as dist
    world = pdist.get_world_size()
    B = Xq.shape[0]
    cand_v = torch.empty((world, B, K), dtype=v.dtype, device=v.device)
    cand_i = torch.empty((world, B, K), dtype=torch.int64, device=v.device)
    dist.all_gather_into_tensor(cand_v.view(-1), v.contiguous().view(-1))
    dist.all_gather_into_tensor(cand_i.view(-1),
                                gidx.contiguous().view(-1))
    allv = cand_v.permute(1, 0, 2).reshape(B, world * K)
    alli = cand_i.permute(1, 0, 2).reshape(B, world * K)
    mv, pos = torch.topk(allv, K, dim=1)
    mi = torch.gather(alli, 1, pos)
    mi = torch.where(mv == float("-inf"),
                     torch.full_like(mi, -1), mi)
    return mv, mi
