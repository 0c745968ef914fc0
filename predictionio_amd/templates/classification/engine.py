"""Classification engine template (Naive Bayes, MI355X-native).

Parity with examples/scala-parallel-classification/add-algorithm/:
- Query {attr0, attr1, attr2} → PredictedResult {label}
  (src/main/scala/Engine.scala)
- DataSource reads `$set` user properties attr0-2 + plan (label)
  (DataSource.scala); readEval k-fold for Accuracy/Precision evaluation
- NaiveBayesAlgorithm: MLlib NaiveBayes.train on LabeledPoints of the 3
  numeric attrs with smoothing lambda (NaiveBayesAlgorithm.scala:35-59)
- Evaluation: Accuracy + Precision metrics (Evaluation.scala,
  PrecisionEvaluation.scala)

This is BASELINE.json config 1 — the CPU end-to-end plumbing config; the
NB math is torch tensor ops (runs on GPU when present, K6 of SURVEY §2.9).
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Any, Dict, List, Optional, Tuple

import torch

from predictionio_amd.controller import (
    Algorithm, AverageMetric, DataSource as BaseDataSource, Engine,
    EngineFactory, Evaluation, MetricEvaluator, OptionAverageMetric,
    Preparator as BasePreparator, SanityCheck, Serving as BaseServing,
)
from predictionio_amd.data import event_store
from predictionio_amd.models.naive_bayes import (
    NaiveBayesModel, train_naive_bayes,
)


@dataclass
class LabeledPoint:
    label: float
    features: List[float]


@dataclass
class TrainingData(SanityCheck):
    labeled_points: List[LabeledPoint]

    def sanity_check(self):
        if not self.labeled_points:
            raise ValueError("no labeled points — check the event store")


@dataclass
class PreparedData:
    labeled_points: List[LabeledPoint]


@dataclass
class Query:
    attr0: float
    attr1: float
    attr2: float

    @staticmethod
    def from_json(d: Dict[str, Any]) -> "Query":
        return Query(float(d["attr0"]), float(d["attr1"]),
                     float(d["attr2"]))


@dataclass
class PredictedResult:
    label: float

    def to_json(self):
        return {"label": self.label}


class DataSource(BaseDataSource):
    """Params: appName, [evalK]."""

    def _read(self) -> List[LabeledPoint]:
        props = event_store.aggregate_properties(
            self.params["appName"], "user",
            required=["plan", "attr0", "attr1", "attr2"])
        return [
            LabeledPoint(float(pm.get("plan")),
                         [float(pm.get("attr0")), float(pm.get("attr1")),
                          float(pm.get("attr2"))])
            for pm in props.values()
        ]

    def read_training(self) -> TrainingData:
        return TrainingData(self._read())

    def read_eval(self):
        k = int(self.params.get("evalK", 5))
        pts = self._read()
        folds = []
        for f in range(k):
            train = [p for i, p in enumerate(pts) if i % k != f]
            test = [p for i, p in enumerate(pts) if i % k == f]
            qa = [(Query(*p.features), p.label) for p in test]
            folds.append((TrainingData(train), {"fold": f}, qa))
        return folds


class Preparator(BasePreparator):
    def prepare(self, td: TrainingData) -> PreparedData:
        return PreparedData(td.labeled_points)


class NaiveBayesAlgorithm(Algorithm):
    """Params: lambda (additive smoothing)."""

    def train(self, pd: PreparedData) -> NaiveBayesModel:
        X = torch.tensor([p.features for p in pd.labeled_points],
                         dtype=torch.float32)
        y = torch.tensor([p.label for p in pd.labeled_points])
        if torch.cuda.is_available():
            X, y = X.cuda(), y.cuda()
        return train_naive_bayes(X, y,
                                 lambda_=float(self.params.get("lambda",
                                                               1.0)))

    def predict(self, model: NaiveBayesModel, query) -> PredictedResult:
        q = query if isinstance(query, Query) else Query.from_json(query)
        x = torch.tensor([q.attr0, q.attr1, q.attr2],
                         device=model.theta.device)
        return PredictedResult(label=float(model.predict(x)))

    def batch_predict(self, model: NaiveBayesModel, queries):
        qs = [(i, q if isinstance(q, Query) else Query.from_json(q))
              for i, q in queries]
        if not qs:
            return []
        X = torch.tensor([[q.attr0, q.attr1, q.attr2] for _, q in qs],
                         device=model.theta.device)
        labels = model.predict(X).cpu()
        return [(i, PredictedResult(label=float(l)))
                for (i, _), l in zip(qs, labels)]


class RandomForestAlgorithm(Algorithm):
    """Random-forest variant (the reference's add-algorithm template adds
    MLlib RandomForest next to NaiveBayes; here scikit-learn). Params:
    numTrees, maxDepth, [seed]."""

    def train(self, pd: PreparedData):
        from sklearn.ensemble import RandomForestClassifier
        X = [p.features for p in pd.labeled_points]
        y = [p.label for p in pd.labeled_points]
        clf = RandomForestClassifier(
            n_estimators=int(self.params.get("numTrees", 10)),
            max_depth=self.params.get("maxDepth"),
            random_state=self.params.get("seed"))
        clf.fit(X, y)
        return clf

    def predict(self, model, query) -> PredictedResult:
        q = query if isinstance(query, Query) else Query.from_json(query)
        label = model.predict([[q.attr0, q.attr1, q.attr2]])[0]
        return PredictedResult(label=float(label))

    def batch_predict(self, model, queries):
        qs = [(i, q if isinstance(q, Query) else Query.from_json(q))
              for i, q in queries]
        if not qs:
            return []
        labels = model.predict(
            [[q.attr0, q.attr1, q.attr2] for _, q in qs])
        return [(i, PredictedResult(label=float(l)))
                for (i, _), l in zip(qs, labels)]


class Serving(BaseServing):
    def serve(self, query, predictions) -> PredictedResult:
        return predictions[0]


class ClassificationEngine(EngineFactory):
    @classmethod
    def apply(cls) -> Engine:
        return Engine(
            data_source_class=DataSource,
            preparator_class=Preparator,
            algorithm_class={"naive": NaiveBayesAlgorithm,
                             "randomforest": RandomForestAlgorithm,
                             "": NaiveBayesAlgorithm},
            serving_class=Serving)


# ------------------------------------------------------------ evaluation

class Accuracy(AverageMetric):
    """Fraction of correct predictions (Evaluation.scala Accuracy)."""

    def calculate_one(self, ei, q, p, a) -> float:
        return 1.0 if p.label == a else 0.0


class Precision(OptionAverageMetric):
    """Precision for one positive label (PrecisionEvaluation.scala):
    counts only queries predicted as `label`."""

    def __init__(self, label: float):
        self.label = label

    def header(self):
        return f"Precision(label={self.label})"

    def calculate_one(self, ei, q, p, a):
        if p.label == self.label:
            return 1.0 if a == self.label else 0.0
        return None  # not predicted positive — excluded


class AccuracyEvaluation(Evaluation):
    """`pio eval` entry (Evaluation.scala)."""

    def __init__(self):
        super().__init__(engine=ClassificationEngine.apply(),
                         metric=Accuracy())
