"""Shared template helpers: device-resident query filters.

The reference's isCandidateItem filters (similarproduct
ALSAlgorithm.scala:245-271, ecommerce ECommAlgorithm.scala:330-430)
re-scan the item metadata dict per query. Here category membership is
precomputed ONCE at train time as per-category uint8 device masks, so
query-time category filtering is a handful of tensor ops — flat in
catalog size."""

from __future__ import annotations

from typing import Dict, Iterable, List, Optional

import torch


class CategoryMasks:
    """Per-category item-membership masks (uint8, 1 = member)."""

    def __init__(self, masks: Dict[str, torch.Tensor], n: int, device):
        self.masks = masks
        self.n = n
        self.device = device

    @classmethod
    def build(cls, items: Dict[str, object], item_map,
              device) -> "CategoryMasks":
        """items: item-id → metadata with a `.categories` list."""
        masks: Dict[str, torch.Tensor] = {}
        n = len(item_map)
        for it, meta in items.items():
            cats = getattr(meta, "categories", None) or []
            if not cats:
                continue
            i = item_map.get(it)
            if i is None:
                continue
            for c in cats:
                m = masks.get(c)
                if m is None:
                    m = torch.zeros(n, dtype=torch.uint8)
                    masks[c] = m
                m[i] = 1
        return cls({c: m.to(device) for c, m in masks.items()}, n, device)

    def banned_outside(self, categories: Iterable[str]) -> torch.Tensor:
        """uint8 mask with 1 for items NOT in any given category — the
        ban contribution of a query's `categories` filter."""
        allowed: Optional[torch.Tensor] = None
        for c in categories:
            m = self.masks.get(c)
            if m is not None:
                allowed = m if allowed is None else (allowed | m)
        if allowed is None:
            return torch.ones(self.n, dtype=torch.uint8,
                              device=self.device)
        return 1 - allowed


def ids_tensor(ids: Iterable[str], id_map, device) -> torch.Tensor:
    """Map string ids through a BiMap into an int64 index tensor,
    dropping unknowns."""
    idxs: List[int] = []
    for s in ids:
        i = id_map.get(s)
        if i is not None:
            idxs.append(i)
    return torch.tensor(idxs, dtype=torch.int64, device=device)
