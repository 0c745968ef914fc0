"""Similar-Product engine template (MI355X-native).

Parity with examples/scala-parallel-similarproduct/recommended-user →
actually the base template (examples/scala-parallel-similarproduct/*):
- Query {items, num, [categories], [whiteList], [blackList]}
  (src/main/scala/Engine.scala)
- DataSource aggregates $set user/item properties + view events
  (DataSource.scala:133 total)
- ALSAlgorithm (P2L): implicit-style ALS on view events with latest-wins
  (user, item) dedup (ALSAlgorithm.scala:88-120), productFeatures collected
  local (:130-143); predict = sum of cosine similarities of the query
  items' factors vs ALL item factors, filtered by white/black/category,
  top-num (:168-271 — the reference's .par loop + hand-rolled cosine +
  bounded PriorityQueue)
- CooccurrenceAlgorithm: user-distinct item pairs → pair counts → top-N
  co-occurring items (CooccurrenceAlgorithm.scala:49-108)

MI355X design: cosine scoring = L2-normalize factors once at train time,
then the summed-query-vector dot against all items in one fused masked
top-K kernel launch.
"""

from __future__ import annotations

import warnings
from collections import defaultdict
from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Set, Tuple

import torch

from predictionio_amd.parallel import dist as pdist

from predictionio_amd.controller import (
    Algorithm, DataSource as BaseDataSource, Engine, EngineFactory,
    Preparator as BasePreparator, SanityCheck, Serving as BaseServing,
)
from predictionio_amd.data import event_store
from predictionio_amd.data.bimap import BiMap
from predictionio_amd.models.als import ALSParams, train_als
from predictionio_amd.ops import topk as topk_ops


@dataclass
class Item:
    categories: Optional[List[str]]


@dataclass
class ViewEvent:
    user: str
    item: str
    t: float


@dataclass
class TrainingData(SanityCheck):
    users: Dict[str, dict]
    items: Dict[str, Item]
    view_events: List[ViewEvent]

    def sanity_check(self):
        if not self.view_events:
            raise ValueError("view events are empty")


@dataclass
class PreparedData:
    users: Dict[str, dict]
    items: Dict[str, Item]
    view_events: List[ViewEvent]


@dataclass
class Query:
    items: List[str]
    num: int
    categories: Optional[List[str]] = None
    white_list: Optional[List[str]] = None
    black_list: Optional[List[str]] = None

    @staticmethod
    def from_json(d: Dict[str, Any]) -> "Query":
        return Query(items=list(d["items"]), num=int(d.get("num", 10)),
                     categories=d.get("categories"),
                     white_list=d.get("whiteList"),
                     black_list=d.get("blackList"))


@dataclass
class ItemScore:
    item: str
    score: float


@dataclass
class PredictedResult:
    item_scores: List[ItemScore]

    def to_json(self):
        return {"itemScores": [{"item": s.item, "score": s.score}
                               for s in self.item_scores]}


class DataSource(BaseDataSource):
    """Params: appName."""

    def read_training(self) -> TrainingData:
        """Params: appName, [eventNames] — the train-with-rate-event
        variant passes ["rate"] (or mixes) and the events still feed the
        implicit trainer as unit preferences."""
        app = self.params["appName"]
        names = list(self.params.get("eventNames", ["view"]))
        users = {eid: pm.to_dict() for eid, pm in
                 event_store.aggregate_properties(app, "user").items()}
        items = {eid: Item(categories=pm.get_opt("categories"))
                 for eid, pm in
                 event_store.aggregate_properties(app, "item").items()}
        views = [
            ViewEvent(e.entity_id, e.target_entity_id,
                      e.event_time.timestamp())
            for e in event_store.find(app, entity_type="user",
                                      event_names=names,
                                      target_entity_type="item")
        ]
        return TrainingData(users, items, views)


class Preparator(BasePreparator):
    def prepare(self, td: TrainingData) -> PreparedData:
        return PreparedData(td.users, td.items, td.view_events)


class SimilarModel:
    """Normalized item factors + maps + categories (P2L local model)."""

    def __init__(self, item_factors_norm: torch.Tensor, item_map: BiMap,
                 items: Dict[str, Item], category_masks=None):
        self.item_factors_norm = item_factors_norm  # L2-normalized rows
        self.item_map = item_map
        self.items = items
        self.item_inv = item_map.inverse_array()
        self.category_masks = category_masks

    def cat_masks(self):
        """Device category masks (built at train time; lazily rebuilt
        for models persisted before round 2)."""
        if self.category_masks is None:
            from predictionio_amd.templates.common import CategoryMasks
            self.category_masks = CategoryMasks.build(
                self.items, self.item_map, self.item_factors_norm.device)
        return self.category_masks


class ALSAlgorithm(Algorithm):
    """Params: rank, numIterations, lambda, alpha, [seed]."""

    def train(self, pd: PreparedData) -> SimilarModel:
        if not pd.view_events:
            raise ValueError("no view events")
        user_map = BiMap.string_int(v.user for v in pd.view_events)
        # include property-only items so queries on them resolve
        item_keys = [v.item for v in pd.view_events] + list(pd.items)
        item_map = BiMap.string_int(item_keys)
        # latest-wins dedup on (user, item) (ALSAlgorithm.scala:88-120):
        # sort by time then aggregate 'latest'; implicit weight 1 per view
        evs = sorted(pd.view_events, key=lambda v: v.t)
        users = torch.tensor([user_map[v.user] for v in evs],
                             dtype=torch.int32)
        items = torch.tensor([item_map[v.item] for v in evs],
                             dtype=torch.int32)
        vals = torch.ones(len(evs))
        from predictionio_amd.ops import als as als_ops
        users, items, vals = als_ops.aggregate_ratings(
            users, items, vals, len(item_map), "sum")
        p = ALSParams(
            rank=int(self.params.get("rank", 10)),
            iterations=int(self.params.get("numIterations", 20)),
            lambda_=float(self.params.get("lambda", 0.01)),
            alpha=float(self.params.get("alpha", 1.0)),
            implicit=True, seed=self.params.get("seed"))
        device = pdist.compute_device()
        _, Y = train_als(users, items, vals, len(user_map), len(item_map),
                         p, device=device)
        Yn = torch.nn.functional.normalize(Y, dim=1, eps=1e-9)
        from predictionio_amd.templates.common import CategoryMasks
        cm = CategoryMasks.build(pd.items, item_map, Yn.device)
        return SimilarModel(Yn, item_map, pd.items, cm)

    def _masks(self, model: SimilarModel, q: Query,
               dev) -> Optional[torch.Tensor]:
        """uint8 banned mask from white/black/category filters
        (ALSAlgorithm.scala:245-271 isCandidateItem). Device-resident:
        category membership comes from train-time masks, white/black
        lists are small index_put ops — nothing scans the catalog."""
        from predictionio_amd.templates.common import ids_tensor
        n = len(model.item_map)
        mask = None
        if q.white_list is not None:
            mask = torch.ones(n, dtype=torch.uint8, device=dev)
            mask[ids_tensor(q.white_list, model.item_map, dev)] = 0
        if q.categories is not None:
            cm = model.cat_masks().banned_outside(q.categories).to(dev)
            mask = cm if mask is None else (mask | cm)
        if q.black_list:
            if mask is None:
                mask = torch.zeros(n, dtype=torch.uint8, device=dev)
            mask[ids_tensor(q.black_list, model.item_map, dev)] = 1
        # query items themselves are never returned (reference excludes them)
        if mask is None:
            mask = torch.zeros(n, dtype=torch.uint8, device=dev)
        mask[ids_tensor(q.items, model.item_map, dev)] = 1
        return mask

    def predict(self, model: SimilarModel, query) -> PredictedResult:
        q = query if isinstance(query, Query) else Query.from_json(query)
        rows = [model.item_map[i] for i in q.items
                if i in model.item_map]
        if not rows:
            return PredictedResult([])
        dev = model.item_factors_norm.device
        # sum of cosines = (sum of normalized query vectors) . Yn
        Xq = model.item_factors_norm[torch.tensor(rows, device=dev)] \
            .sum(dim=0, keepdim=True)
        mask = self._masks(model, q, dev)
        v, idx = topk_ops.topk_score(Xq, model.item_factors_norm, q.num,
                                     item_mask=mask)
        v, idx = v[0].cpu(), idx[0].cpu()
        return PredictedResult([
            ItemScore(model.item_inv[int(i)], float(s))
            for s, i in zip(v, idx) if i >= 0])


class CooccurrenceModel:
    def __init__(self, top_n: Dict[int, List[Tuple[int, int]]],
                 item_map: BiMap, items: Dict[str, Item]):
        self.top_n = top_n          # item → [(other_item, count)]
        self.item_map = item_map
        self.items = items
        self.item_inv = item_map.inverse_array()


class CooccurrenceAlgorithm(Algorithm):
    """Params: n (top co-occurrences kept per item)
    (CooccurrenceAlgorithm.scala:49-108)."""

    def train(self, pd: PreparedData) -> CooccurrenceModel:
        """Pair counting as ONE sparse Gramian: C = A^T A with A the
        user×item view-incidence matrix (user-distinct), then a
        vectorized per-item top-N — replaces the reference's self-join
        + groupBy (CooccurrenceAlgorithm.scala:55-108) and round 1's
        Python pair loops (quadratic in per-user views). Runs on the
        GPU when available (torch sparse mm)."""
        import numpy as np
        item_map = BiMap.string_int(
            [v.item for v in pd.view_events] + list(pd.items))
        user_map = BiMap.string_int(v.user for v in pd.view_events)
        n_keep = int(self.params.get("n", 10))
        n_items = len(item_map)
        n_users = len(user_map)
        if not pd.view_events:
            return CooccurrenceModel({}, item_map, pd.items)
        u = torch.tensor([user_map[v.user] for v in pd.view_events],
                         dtype=torch.int64)
        i = torch.tensor([item_map[v.item] for v in pd.view_events],
                         dtype=torch.int64)
        # user-distinct (view twice ≠ two pairs)
        key = u * n_items + i
        key = torch.unique(key)
        u, i = key // n_items, key % n_items
        device = (torch.device("cuda") if torch.cuda.is_available()
                  else torch.device("cpu"))
        A = torch.sparse_coo_tensor(
            torch.stack([u, i]).to(device),
            torch.ones(u.numel(), device=device),
            (n_users, n_items)).coalesce()
        with warnings.catch_warnings():
            # torch's own "sparse CSR is beta" notice, not actionable
            warnings.simplefilter("ignore", UserWarning)
            C = torch.sparse.mm(A.t(), A).coalesce()
        ii = C.indices()[0].cpu().numpy()
        jj = C.indices()[1].cpu().numpy()
        vv = C.values().cpu().numpy()
        off = ii != jj  # drop the diagonal (item with itself)
        ii, jj, vv = ii[off], jj[off], vv[off].astype(np.int64)
        # per-item top-N: lexsort by (item, -count), take first n_keep
        order = np.lexsort((-vv, ii))
        ii, jj, vv = ii[order], jj[order], vv[order]
        starts = np.flatnonzero(np.r_[True, ii[1:] != ii[:-1]])
        top_n: Dict[int, List[Tuple[int, int]]] = {}
        bounds = np.r_[starts, len(ii)]
        for s, e in zip(bounds[:-1], bounds[1:]):
            e2 = min(e, s + n_keep)
            top_n[int(ii[s])] = list(zip(jj[s:e2].tolist(),
                                         vv[s:e2].tolist()))
        return CooccurrenceModel(top_n, item_map, pd.items)

    def predict(self, model: CooccurrenceModel, query) -> PredictedResult:
        q = query if isinstance(query, Query) else Query.from_json(query)
        counts: Dict[int, int] = defaultdict(int)
        qidx = {model.item_map.get(i) for i in q.items}
        for i in q.items:
            ii = model.item_map.get(i)
            if ii is None:
                continue
            for other, c in model.top_n.get(ii, []):
                counts[other] += c
        white = ({model.item_map.get(i) for i in q.white_list}
                 if q.white_list is not None else None)
        black = {model.item_map.get(i) for i in (q.black_list or [])}
        cats = set(q.categories) if q.categories is not None else None

        def ok(i: int) -> bool:
            if i in qidx or i in black:
                return False
            if white is not None and i not in white:
                return False
            if cats is not None:
                meta = model.items.get(model.item_inv[i])
                if not (meta and meta.categories
                        and cats & set(meta.categories)):
                    return False
            return True

        ranked = sorted(((c, i) for i, c in counts.items() if ok(i)),
                        reverse=True)[:q.num]
        return PredictedResult([
            ItemScore(model.item_inv[i], float(c)) for c, i in ranked])


class Serving(BaseServing):
    def serve(self, query, predictions) -> PredictedResult:
        return predictions[0]


class SimilarProductEngine(EngineFactory):
    @classmethod
    def apply(cls) -> Engine:
        return Engine(
            data_source_class=DataSource,
            preparator_class=Preparator,
            algorithm_class={"als": ALSAlgorithm,
                             "cooccurrence": CooccurrenceAlgorithm,
                             "": ALSAlgorithm},
            serving_class=Serving)


# --------------------------------------------------------------------------
# recommended-user variant (examples/scala-parallel-similarproduct/
# recommended-user): the same cosine-kNN machinery over USER factors —
# "users who follow the query users also follow these users". Events are
# `follow` (user → similarUser); the model keeps the followed-user factors.
# --------------------------------------------------------------------------


@dataclass
class UserQuery:
    users: List[str]
    num: int
    white_list: Optional[List[str]] = None
    black_list: Optional[List[str]] = None

    @staticmethod
    def from_json(d: Dict[str, Any]) -> "UserQuery":
        return UserQuery(users=list(d["users"]), num=int(d.get("num", 10)),
                         white_list=d.get("whiteList"),
                         black_list=d.get("blackList"))


class FollowDataSource(BaseDataSource):
    """Reads `follow` events (user → similarUser entity type `user`)."""

    def read_training(self) -> TrainingData:
        app = self.params["appName"]
        follows = [
            ViewEvent(e.entity_id, e.target_entity_id,
                      e.event_time.timestamp())
            for e in event_store.find(app, entity_type="user",
                                      event_names=["follow"],
                                      target_entity_type="user")
        ]
        return TrainingData({}, {}, follows)


class RecommendedUserAlgorithm(ALSAlgorithm):
    """Trains implicit ALS on follow events; the 'item' side is the
    followed users, so predict returns similar USERS
    (recommended-user ALSAlgorithm.scala)."""

    def predict(self, model: SimilarModel, query) -> PredictedResult:
        q = query if isinstance(query, UserQuery) \
            else UserQuery.from_json(query)
        return super().predict(
            model, Query(items=q.users, num=q.num,
                         white_list=q.white_list, black_list=q.black_list))


class RecommendedUserEngine(EngineFactory):
    @classmethod
    def apply(cls) -> Engine:
        return Engine(
            data_source_class=FollowDataSource,
            preparator_class=Preparator,
            algorithm_class={"als": RecommendedUserAlgorithm,
                             "": RecommendedUserAlgorithm},
            serving_class=Serving)
