"""E-Commerce Recommendation engine template (MI355X-native).

Parity with examples/scala-parallel-ecommercerecommendation/adjust-score/
ECommAlgorithm.scala (649 LoC):
- train: implicit ALS on view + buy events (:123-133) with buy weighted
  over view, plus popularity counts for the default path
  (trainDefault :207-246)
- predict consults the LIVE event store at query time:
  * genBlackList (:330-398): query blackList + the user's seen events
    (unseenOnly) + `$set constraint unavailableItems` — each lookup under
    the 200 ms timeout convention (:341, :377, :445), degrading to empty
    on timeout
  * known user → predictKnownUser (:471-506): dot products over all item
    factors, masked, top-num
  * unknown/cold user → getRecentItems (:430-467) → predictSimilar
    (:541-599): cosine vs recent view items; if none → predictDefault
    (:508-539): popularity ranking
  * optional weightedItems score multipliers (:400-430, adjust-score
    variant)

MI355X design: all three scoring paths are fused masked top-K kernel
launches over device-resident factors.
"""

from __future__ import annotations

from collections import defaultdict
from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Set

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
class ItemEvent:
    user: str
    item: str
    t: float


@dataclass
class TrainingData(SanityCheck):
    users: Dict[str, dict]
    items: Dict[str, Item]
    view_events: List[ItemEvent]
    buy_events: List[ItemEvent]
    view_columns: Optional["EventColumns"] = None
    buy_columns: Optional["EventColumns"] = None

    def sanity_check(self):
        if not self.view_events and not (
                self.view_columns and len(self.view_columns)):
            raise ValueError("view events are empty")


@dataclass
class EventColumns:
    """Columnar (user, item) event arrays from the bulk store read."""
    users: Any   # np.ndarray object
    items: Any   # np.ndarray object

    def __len__(self):
        return len(self.users)


@dataclass
class PreparedData:
    users: Dict[str, dict]
    items: Dict[str, Item]
    view_events: List[ItemEvent]
    buy_events: List[ItemEvent]
    view_columns: Optional[EventColumns] = None
    buy_columns: Optional[EventColumns] = None


@dataclass
class WeightGroup:
    items: Set[str]
    weight: float


@dataclass
class Query:
    user: str
    num: int
    categories: Optional[List[str]] = None
    white_list: Optional[List[str]] = None
    black_list: Optional[List[str]] = None

    @staticmethod
    def from_json(d: Dict[str, Any]) -> "Query":
        return Query(user=d["user"], num=int(d.get("num", 10)),
                     categories=d.get("categories"),
                     white_list=d.get("whiteList"),
                     black_list=d.get("blackList"))


@dataclass
class ItemScore:
    item: str
    score: float

    def to_json(self):
        return {"item": self.item, "score": self.score}


@dataclass
class PredictedResult:
    item_scores: List[ItemScore]

    def to_json(self):
        return {"itemScores": [s.to_json() for s in self.item_scores]}


class DataSource(BaseDataSource):
    """Params: appName."""

    def read_training(self) -> TrainingData:
        app = self.params["appName"]
        users = {eid: pm.to_dict() for eid, pm in
                 event_store.aggregate_properties(app, "user").items()}
        items = {eid: Item(categories=pm.get_opt("categories"))
                 for eid, pm in
                 event_store.aggregate_properties(app, "item").items()}

        if self.params.get("columnar", True):
            import numpy as np

            def read_cols(name):
                cols = event_store.find_columns(
                    app, entity_type="user", event_names=[name],
                    target_entity_type="item")
                return EventColumns(
                    users=np.asarray(cols["entity_id"], dtype=object),
                    items=np.asarray(cols["target_entity_id"],
                                     dtype=object))

            return TrainingData(users, items, [], [],
                                view_columns=read_cols("view"),
                                buy_columns=read_cols("buy"))

        def read(name):
            return [ItemEvent(e.entity_id, e.target_entity_id,
                              e.event_time.timestamp())
                    for e in event_store.find(app, entity_type="user",
                                              event_names=[name],
                                              target_entity_type="item")]

        return TrainingData(users, items, read("view"), read("buy"))


class Preparator(BasePreparator):
    def prepare(self, td: TrainingData) -> PreparedData:
        return PreparedData(td.users, td.items, td.view_events,
                            td.buy_events, td.view_columns,
                            td.buy_columns)


class ECommModel:
    def __init__(self, rank: int, user_features: torch.Tensor,
                 item_factors: torch.Tensor, item_factors_norm: torch.Tensor,
                 user_map: BiMap, item_map: BiMap,
                 items: Dict[str, Item], popular_count: Dict[int, int],
                 category_masks=None):
        self.rank = rank
        self.user_features = user_features
        self.item_factors = item_factors
        self.item_factors_norm = item_factors_norm
        self.user_map = user_map
        self.item_map = item_map
        self.items = items
        self.popular_count = popular_count
        self.item_inv = item_map.inverse_array()
        self.category_masks = category_masks

    def cat_masks(self):
        """Device category masks (train-time; lazily rebuilt for models
        persisted before round 2)."""
        if self.category_masks is None:
            from predictionio_amd.templates.common import CategoryMasks
            self.category_masks = CategoryMasks.build(
                self.items, self.item_map, self.item_factors.device)
        return self.category_masks


class ECommAlgorithm(Algorithm):
    """Params: appName, unseenOnly, seenEvents, similarEvents, rank,
    numIterations, lambda, alpha, [seed], [buyScore]."""

    def train(self, pd: PreparedData) -> ECommModel:
        # Reference parity: ECommAlgorithm.genMLlibRating builds MLlib
        # ratings from viewEvents ONLY (ECommAlgorithm.scala:168-205);
        # buys feed just the popularity counts. buyScore>0 blends buys
        # into training as an opt-in extension beyond the reference.
        buy_score = float(self.params.get("buyScore", 0.0))
        import numpy as np
        from predictionio_amd.ops import als as als_ops
        if pd.view_columns is not None:
            # columnar ingest: one joint factorize per id space gives the
            # BiMaps AND the event codes in a single C pass
            import pandas as pandas_
            vc, bc = pd.view_columns, pd.buy_columns
            nv, nb = len(vc), len(bc) if bc is not None else 0
            ucat = np.concatenate(
                [vc.users, bc.users if bc is not None else [],
                 np.asarray(list(pd.users), dtype=object)])
            icat = np.concatenate(
                [vc.items, bc.items if bc is not None else [],
                 np.asarray(list(pd.items), dtype=object)])
            ucodes, uu = pandas_.factorize(ucat)
            icodes, ui = pandas_.factorize(icat)
            user_map = BiMap.from_uniques(uu)
            item_map = BiMap.from_uniques(ui)
            if buy_score > 0.0 and nb:
                users_np = ucodes[:nv + nb]
                items_np = icodes[:nv + nb]
                vals_np = np.concatenate(
                    [np.ones(nv, np.float32),
                     np.full(nb, buy_score, np.float32)])
            else:
                users_np = ucodes[:nv]
                items_np = icodes[:nv]
                vals_np = np.ones(nv, np.float32)
            users = torch.from_numpy(users_np.astype(np.int32))
            items = torch.from_numpy(items_np.astype(np.int32))
            vals = torch.from_numpy(vals_np)
            buy_item_codes = icodes[nv:nv + nb]
        else:
            user_map = BiMap.string_int(
                [e.user for e in pd.view_events]
                + [e.user for e in pd.buy_events] + list(pd.users))
            item_map = BiMap.string_int(
                [e.item for e in pd.view_events]
                + [e.item for e in pd.buy_events] + list(pd.items))
            evs = ([(e, 1.0) for e in pd.view_events]
                   + ([(e, buy_score) for e in pd.buy_events]
                      if buy_score > 0.0 else []))
            users = torch.tensor([user_map[e.user] for e, _ in evs],
                                 dtype=torch.int32)
            items = torch.tensor([item_map[e.item] for e, _ in evs],
                                 dtype=torch.int32)
            vals = torch.tensor([w for _, w in evs], dtype=torch.float32)
            buy_item_codes = None
        users, items, vals = als_ops.aggregate_ratings(
            users, items, vals, len(item_map), "sum")
        p = ALSParams(
            rank=int(self.params.get("rank", 10)),
            iterations=int(self.params.get("numIterations", 20)),
            lambda_=float(self.params.get("lambda", 0.01)),
            alpha=float(self.params.get("alpha", 1.0)),
            implicit=True, seed=self.params.get("seed"))
        device = pdist.compute_device()
        X, Y = train_als(users, items, vals, len(user_map), len(item_map),
                         p, device=device)
        # popularity = buy counts (trainDefault :207-246)
        pop: Dict[int, int] = defaultdict(int)
        if buy_item_codes is not None:
            uniq, cnt = np.unique(buy_item_codes, return_counts=True)
            pop = dict(zip(uniq.tolist(), cnt.tolist()))
        else:
            for e in pd.buy_events:
                pop[item_map[e.item]] += 1
        Yn = torch.nn.functional.normalize(Y, dim=1, eps=1e-9)
        from predictionio_amd.templates.common import CategoryMasks
        cm = CategoryMasks.build(pd.items, item_map, Y.device)
        return ECommModel(p.rank, X, Y, Yn, user_map, item_map, pd.items,
                          dict(pop), cm)

    # ------------------------------------------------------------ filters

    def _gen_black_list(self, q: Query) -> Set[str]:
        """Query blacklist + seen events + unavailable items, each live
        lookup bounded by 200 ms (genBlackList :330-398)."""
        black: Set[str] = set(q.black_list or [])
        if self.params.get("unseenOnly", False):
            seen_events = self.params.get("seenEvents", ["view", "buy"])
            try:
                seen = event_store.find_by_entity(
                    app_name=self.params["appName"], entity_type="user",
                    entity_id=q.user, event_names=seen_events,
                    target_entity_type="item", timeout=0.2)
                black |= {e.target_entity_id for e in seen}
            except TimeoutError:
                pass  # degrade: no seen filter (reference logs + continues)
        try:
            constr = event_store.find_by_entity(
                app_name=self.params["appName"], entity_type="constraint",
                entity_id="unavailableItems", event_names=["$set"],
                limit=1, latest=True, timeout=0.2)
            if constr:
                black |= set(constr[0].properties.get("items") or [])
        except TimeoutError:
            pass
        return black

    def _mask(self, model: ECommModel, q: Query, black: Set[str]
              ) -> torch.Tensor:
        """Device-resident filters: category membership from train-time
        masks, white/black lists as small index_put ops — query cost is
        flat in catalog size."""
        from predictionio_amd.templates.common import ids_tensor
        dev = model.item_factors.device
        n = len(model.item_map)
        mask = torch.zeros(n, dtype=torch.uint8, device=dev)
        if q.white_list is not None:
            mask[:] = 1
            mask[ids_tensor(q.white_list, model.item_map, dev)] = 0
        if q.categories is not None:
            mask |= model.cat_masks().banned_outside(q.categories).to(dev)
        mask[ids_tensor(black, model.item_map, dev)] = 1
        return mask

    def _weight_groups(self) -> List[WeightGroup]:
        """adjust-score variant: [{"items": [...], "weight": w}]
        (weightedItems :400-430)."""
        return [WeightGroup(set(g["items"]), float(g["weight"]))
                for g in self.params.get("weightedItems", [])]

    def _apply_weights(self, model, scores: List[ItemScore]
                       ) -> List[ItemScore]:
        groups = self._weight_groups()
        if not groups:
            return scores
        out = []
        for s in scores:
            w = 1.0
            for g in groups:
                if s.item in g.items:
                    w *= g.weight
            out.append(ItemScore(s.item, s.score * w))
        out.sort(key=lambda s: -s.score)
        return out

    # ------------------------------------------------------------ predict

    def _topk(self, model, Xq, mask, num) -> List[ItemScore]:
        v, idx = topk_ops.topk_score(Xq, model.item_factors, num,
                                     item_mask=mask)
        v, idx = v[0].cpu(), idx[0].cpu()
        return [ItemScore(model.item_inv[int(i)], float(s))
                for s, i in zip(v, idx) if i >= 0]

    def _topk_cos(self, model, Xq, mask, num) -> List[ItemScore]:
        v, idx = topk_ops.topk_score(Xq, model.item_factors_norm, num,
                                     item_mask=mask)
        v, idx = v[0].cpu(), idx[0].cpu()
        return [ItemScore(model.item_inv[int(i)], float(s))
                for s, i in zip(v, idx) if i >= 0]

    def _recent_items(self, q: Query) -> List[str]:
        """Latest N similarEvents of the user (getRecentItems :430-467)."""
        try:
            evs = event_store.find_by_entity(
                app_name=self.params["appName"], entity_type="user",
                entity_id=q.user,
                event_names=self.params.get("similarEvents", ["view"]),
                target_entity_type="item", limit=10, latest=True,
                timeout=0.2)
            return [e.target_entity_id for e in evs]
        except TimeoutError:
            return []

    def predict(self, model: ECommModel, query) -> PredictedResult:
        q = query if isinstance(query, Query) else Query.from_json(query)
        black = self._gen_black_list(q)
        mask = self._mask(model, q, black)
        uidx = model.user_map.get(q.user)
        if uidx is not None and uidx < model.user_features.shape[0]:
            scores = self._topk(model, model.user_features[uidx:uidx + 1],
                                mask, q.num)
        else:
            recent = [model.item_map[i] for i in self._recent_items(q)
                      if i in model.item_map]
            if recent:
                dev = model.item_factors_norm.device
                Xq = model.item_factors_norm[
                    torch.tensor(recent, device=dev)].sum(dim=0,
                                                          keepdim=True)
                scores = self._topk_cos(model, Xq, mask, q.num)
            else:
                scores = self._predict_default(model, mask, q)
        return PredictedResult(self._apply_weights(model, scores)[:q.num])

    def _predict_default(self, model: ECommModel, mask, q: Query
                         ) -> List[ItemScore]:
        """Popularity ranking under the same filters (:508-539)."""
        m = mask.cpu()
        ranked = sorted(
            ((c, i) for i, c in model.popular_count.items()
             if not bool(m[i])), reverse=True)[:q.num]
        return [ItemScore(model.item_inv[i], float(c)) for c, i in ranked]


class Serving(BaseServing):
    def serve(self, query, predictions) -> PredictedResult:
        return predictions[0]


class ECommerceRecommendationEngine(EngineFactory):
    @classmethod
    def apply(cls) -> Engine:
        return Engine(
            data_source_class=DataSource,
            preparator_class=Preparator,
            algorithm_class={"ecomm": ECommAlgorithm, "": ECommAlgorithm},
            serving_class=Serving)
