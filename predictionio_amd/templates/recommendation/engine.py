"""Recommendation engine template (MLlib-ALS template, MI355X-native).

Parity with examples/scala-parallel-recommendation/blacklist-items/:
- Query {user, num, [blackList]} → PredictedResult {itemScores}
  (src/main/scala/Engine.scala:25-40; blacklist variant)
- DataSource reads `rate` + `buy` events → Ratings (DataSource.scala:45-80);
  readEval k-fold split for evaluation (:83-105)
- ALSAlgorithm: BiMap ID compaction (ALSAlgorithm.scala:60-61), explicit
  ALS (:75-86) or implicit (train-with-view variant `trainImplicit`);
  eventTime-ordered dedup keeps the LATEST rating per (user, item) for
  explicit data (MLlibRating aggregation in the template)
- ALSModel: factor matrices + BiMaps, custom persistence
  (ALSModel.scala:62-100); predict = recommendProductsWithFilter — dot
  products over all items, blacklist filtered, top-num (ALSModel.scala:44-60)
- batchPredict for evaluation (ALSAlgorithm.scala:117-158)

MI355X design: the per-item `blas.ddot` loop + `top(num)` becomes one
fused masked top-K HIP kernel launch (ops.topk.topk_score); training is
the fused Gramian+Cholesky/Woodbury solver (ops.als) instead of MLlib's
shuffle-based ALS.
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple

import torch

from predictionio_amd.parallel import dist as pdist

from predictionio_amd.controller import (
    Algorithm, DataSource as BaseDataSource, Engine, EngineFactory, Params,
    PersistentModel, Preparator as BasePreparator, SanityCheck, Serving as
    BaseServing,
)
from predictionio_amd.data import event_store
from predictionio_amd.data.bimap import BiMap
from predictionio_amd.models.als import ALSParams, train_als
from predictionio_amd.ops import topk as topk_ops


@dataclass
class Rating:
    user: str
    item: str
    rating: float


@dataclass
class RatingColumns:
    """Columnar rating triples from the bulk event-store read
    (event_store.find_columns) — numpy id arrays + fp32 ratings, no
    per-event Python objects. The training ingest path for 10^7+ event
    stores."""
    users: Any   # np.ndarray of user ids
    items: Any   # np.ndarray of item ids
    ratings: Any  # np.ndarray float32

    def __len__(self):
        return len(self.ratings)


@dataclass
class TrainingData(SanityCheck):
    ratings: List[Rating] = field(default_factory=list)
    columns: Optional[RatingColumns] = None

    def sanity_check(self):
        if not self.ratings and not (self.columns and len(self.columns)):
            raise ValueError("ratings is empty — check the event store")


@dataclass
class PreparedData:
    ratings: List[Rating] = field(default_factory=list)
    columns: Optional[RatingColumns] = None


@dataclass
class Query:
    user: str
    num: int
    black_list: Optional[List[str]] = None

    @staticmethod
    def from_json(d: Dict[str, Any]) -> "Query":
        return Query(user=d["user"], num=int(d.get("num", 10)),
                     black_list=d.get("blackList"))


@dataclass
class ItemScore:
    item: str
    score: float


@dataclass
class PredictedResult:
    item_scores: List[ItemScore]

    def to_json(self) -> Dict[str, Any]:
        return {"itemScores": [{"item": s.item, "score": s.score}
                               for s in self.item_scores]}


class DataSource(BaseDataSource):
    """Reads rate/buy events (DataSource.scala:45-80). Params: appName,
    [evalParams {kFold, queryNum}]."""

    def _read(self) -> List[Rating]:
        """Reads `eventNames` (default rate+buy; the train-with-view
        variant passes ["view"]); events without a rating property score
        `implicitRating` (the reference maps buy → 4.0)."""
        names = self.params.get("eventNames", ["rate", "buy"])
        implicit_r = float(self.params.get("implicitRating", 4.0))
        events = event_store.find(
            app_name=self.params["appName"],
            entity_type="user", event_names=list(names),
            target_entity_type="item")
        ratings = []
        for e in events:
            if e.event == "rate" and "rating" in e.properties:
                r = float(e.properties.get("rating"))
            else:  # view/buy events → implicit preference weight
                r = implicit_r
            ratings.append(Rating(e.entity_id, e.target_entity_id, r))
        return ratings

    def _read_columns(self) -> RatingColumns:
        """Columnar bulk read: one store-side scan + vectorized rating
        derivation (rate events use their rating property, others the
        implicit weight) — the event-store→device ingest path measured
        in scripts/ingest_bench.py."""
        import numpy as np
        names = self.params.get("eventNames", ["rate", "buy"])
        implicit_r = float(self.params.get("implicitRating", 4.0))
        cols = event_store.find_columns(
            app_name=self.params["appName"], entity_type="user",
            event_names=list(names), target_entity_type="item",
            property_fields=["rating"])
        ev = np.asarray(cols["event"], dtype=object)
        users = np.asarray(cols["entity_id"], dtype=object)
        items = np.asarray(cols["target_entity_id"], dtype=object)
        rat = np.asarray(cols["rating"], dtype=object)
        vals = np.full(len(ev), implicit_r, dtype=np.float32)
        mask = (ev == "rate") & (rat != None)  # noqa: E711 (elementwise)
        if mask.any():
            vals[mask] = rat[mask].astype(np.float32)
        return RatingColumns(users=users, items=items, ratings=vals)

    def read_training(self) -> TrainingData:
        if self.params.get("columnar", True):
            return TrainingData(columns=self._read_columns())
        return TrainingData(self._read())

    def read_eval(self):
        """k-fold split (DataSource.scala:83-105): fold k tests on
        elements with index % kFold == k, trains on the rest. Actuals are
        the items each user rated in the test split."""
        ep = self.params.get("evalParams") or {}
        k_fold = int(ep.get("kFold", 5))
        query_num = int(ep.get("queryNum", 10))
        ratings = self._read()
        folds = []
        for k in range(k_fold):
            train = [r for i, r in enumerate(ratings) if i % k_fold != k]
            test = [r for i, r in enumerate(ratings) if i % k_fold == k]
            by_user: Dict[str, List[str]] = {}
            for r in test:
                by_user.setdefault(r.user, []).append(r.item)
            qa = [(Query(user=u, num=query_num), items)
                  for u, items in by_user.items()]
            folds.append((TrainingData(train), {"fold": k}, qa))
        return folds


class Preparator(BasePreparator):
    def prepare(self, td: TrainingData) -> PreparedData:
        return PreparedData(td.ratings, td.columns)


class ALSModel(PersistentModel):
    """Factors + ID maps; persisted as .pt shards (the reference persists
    MatrixFactorizationModel object files + BiMaps, ALSModel.scala:62-100)."""

    def __init__(self, rank: int, user_features: torch.Tensor,
                 product_features: torch.Tensor, user_map: BiMap,
                 item_map: BiMap):
        self.rank = rank
        self.user_features = user_features
        self.product_features = product_features
        self.user_map = user_map
        self.item_map = item_map
        self._item_inv: Optional[List[str]] = None

    @property
    def item_inv(self) -> List[str]:
        if self._item_inv is None:
            self._item_inv = self.item_map.inverse_array()
        return self._item_inv

    def item_inv_np(self):
        """item id -> external string as a numpy array (vectorized
        fancy-indexing for the batch-predict epilogue)."""
        if getattr(self, "_item_inv_np", None) is None:
            import numpy as np
            self._item_inv_np = np.asarray(self.item_inv, dtype=object)
        return self._item_inv_np

    @staticmethod
    def _dir(instance_id: str) -> str:
        base = os.environ.get("PIO_FS_BASEDIR",
                              os.path.expanduser("~/.pio_store"))
        return os.path.join(base, "models", f"als-{instance_id}")

    def save(self, instance_id: str, params: Params) -> bool:
        d = self._dir(instance_id)
        os.makedirs(d, exist_ok=True)
        torch.save({
            "rank": self.rank,
            "user_features": self.user_features.cpu(),
            "product_features": self.product_features.cpu(),
            "user_map": self.user_map.to_dict(),
            "item_map": self.item_map.to_dict(),
        }, os.path.join(d, "model.pt"))
        return True

    @classmethod
    def load(cls, instance_id: str, params: Params) -> "ALSModel":
        blob = torch.load(os.path.join(cls._dir(instance_id), "model.pt"),
                          weights_only=False)
        m = cls(blob["rank"], blob["user_features"],
                blob["product_features"], BiMap(blob["user_map"]),
                BiMap(blob["item_map"]))
        if torch.cuda.is_available():
            m.user_features = m.user_features.cuda()
            m.product_features = m.product_features.cuda()
        return m


class ALSAlgorithm(Algorithm):
    """Params: rank, numIterations, lambda, [alpha], [implicitPrefs],
    [seed] (engine.json `als` params in the reference template)."""

    def train(self, pd: PreparedData) -> ALSModel:
        if pd.columns is not None and len(pd.columns):
            # columnar ingest: C-speed ID compaction (pandas.factorize —
            # first-seen codes, same assignment as BiMap.string_int) and
            # zero-copy tensor construction
            import numpy as np
            import pandas as pandas_
            c = pd.columns
            cu, uu = pandas_.factorize(c.users)
            ci, ui = pandas_.factorize(c.items)
            user_map = BiMap.from_uniques(uu)
            item_map = BiMap.from_uniques(ui)
            users = torch.from_numpy(cu.astype(np.int32))
            items = torch.from_numpy(ci.astype(np.int32))
            vals = torch.from_numpy(np.ascontiguousarray(c.ratings))
        elif not pd.ratings:
            raise ValueError("empty ratings")
        else:
            ratings = pd.ratings
            user_map = BiMap.string_int(r.user for r in ratings)
            item_map = BiMap.string_int(r.item for r in ratings)
            users = torch.tensor([user_map[r.user] for r in ratings],
                                 dtype=torch.int32)
            items = torch.tensor([item_map[r.item] for r in ratings],
                                 dtype=torch.int32)
            vals = torch.tensor([r.rating for r in ratings],
                                dtype=torch.float32)
        implicit = bool(self.params.get("implicitPrefs", False))
        from predictionio_amd.ops import als as als_ops
        # dedup semantics: explicit keeps the LATEST rating per pair
        # (events are time-ordered), implicit sums weights
        users, items, vals = als_ops.aggregate_ratings(
            users, items, vals, len(item_map),
            "sum" if implicit else "latest")
        p = ALSParams(
            rank=int(self.params.get("rank", 10)),
            iterations=int(self.params.get("numIterations", 10)),
            lambda_=float(self.params.get("lambda", 0.01)),
            alpha=float(self.params.get("alpha", 1.0)),
            implicit=implicit,
            seed=self.params.get("seed"),
            # reference checkpoints ALS every 10 iterations
            # (ALSAlgorithm.scala:85); here it is crash-resume
            checkpoint_every=int(self.params.get("checkpointEvery", 0)),
            checkpoint_dir=self.params.get("checkpointDir"))
        device = pdist.compute_device()
        X, Y = train_als(users, items, vals, len(user_map), len(item_map),
                         p, device=device)
        return ALSModel(p.rank, X, Y, user_map, item_map)

    def predict(self, model: ALSModel, query) -> PredictedResult:
        q = query if isinstance(query, Query) else Query.from_json(query)
        uidx = model.user_map.get(q.user)
        if uidx is None:
            return PredictedResult([])  # unseen user
        bl = None
        if q.black_list:
            banned = sorted(model.item_map[b] for b in q.black_list
                            if b in model.item_map)
            if banned:
                bl = (torch.tensor([0, len(banned)], dtype=torch.int64),
                      torch.tensor(banned, dtype=torch.int32))
        Xq = model.user_features[uidx:uidx + 1]
        dev = Xq.device
        v, idx = topk_ops.topk_score(
            Xq, model.product_features, q.num,
            ban_indptr=bl[0].to(dev) if bl else None,
            ban_indices=bl[1].to(dev) if bl else None)
        v, idx = v[0].cpu(), idx[0].cpu()
        inv = model.item_inv
        scores = [ItemScore(inv[int(i)], float(s))
                  for s, i in zip(v, idx) if i >= 0]
        return PredictedResult(scores)

    def batch_predict(self, model: ALSModel, queries):
        """Device-batched: one fused top-K launch for all queries
        (replaces the reference's cartesian+groupBy batch path,
        ALSAlgorithm.scala:117-158)."""
        idxs, rows = [], []
        for i, q in queries:
            qq = q if isinstance(q, Query) else Query.from_json(q)
            u = model.user_map.get(qq.user)
            if u is not None:
                idxs.append((i, qq))
                rows.append(u)
        out: Dict[int, PredictedResult] = {
            i: PredictedResult([]) for i, _ in queries}
        if rows:
            Xq = model.user_features[torch.tensor(rows)]
            num = max(q.num for _, q in idxs)
            v, ix = topk_ops.topk_score(Xq, model.product_features, num)
            # vectorized epilogue: per-element .item() calls and list
            # lookups made the Python side ~20x the kernel cost at 50k
            # queries (profiles/batchpredict_r2.log)
            import numpy as np
            v_np = v.cpu().numpy()
            ix_np = ix.cpu().numpy()
            names = model.item_inv_np()[np.clip(ix_np, 0, None)]
            for r, (i, qq) in enumerate(idxs):
                n = qq.num
                out[i] = PredictedResult(
                    [ItemScore(nm, float(s)) for nm, s, it in
                     zip(names[r, :n], v_np[r, :n], ix_np[r, :n])
                     if it >= 0])
        return list(out.items())


class Serving(BaseServing):
    def serve(self, query, predictions: List[PredictedResult]
              ) -> PredictedResult:
        return predictions[0]


class RecommendationEngine(EngineFactory):
    @classmethod
    def apply(cls) -> Engine:
        return Engine(
            data_source_class=DataSource,
            preparator_class=Preparator,
            algorithm_class={"als": ALSAlgorithm, "": ALSAlgorithm},
            serving_class=Serving)
