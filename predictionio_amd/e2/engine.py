"""e2 engine library: CategoricalNaiveBayes, BinaryVectorizer, MarkovChain,
CrossValidation.

Parity with the reference e2 module:
- CategoricalNaiveBayes.train: combineByKey label/feature counting →
  log priors/likelihoods; model logScore/predict
  (e2/.../engine/CategoricalNaiveBayes.scala:29-79, 101-170)
- BinaryVectorizer: categorical (field, value) → one-hot double array
  (BinaryVectorizer.scala:26-63)
- MarkovChain: sparse row-normalized transition matrix + topNProbs
  (MarkovChain.scala:28-87)
- CrossValidation: k-fold split by index % k (CrossValidation.scala:33-67)

The reference builds these on Spark RDD combinators; here the counting is
plain-Python/torch over in-memory sequences (the event volumes these serve
are metadata-scale; the GPU hot paths live in predictionio_amd.ops).
"""

from __future__ import annotations

import math
from collections import defaultdict
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

import torch


# ------------------------------------------------------- naive bayes (e2)

@dataclass
class LabeledPoint:
    """(label, categorical feature values) (CategoricalNaiveBayes.scala)."""
    label: str
    features: Sequence[str]


@dataclass
class CategoricalNaiveBayesModel:
    """Log priors + per-(feature-position, value) log likelihoods
    (CategoricalNaiveBayes.scala:101-170)."""
    priors: Dict[str, float]
    likelihoods: Dict[str, List[Dict[str, float]]]

    def log_score(self, point: LabeledPoint,
                  default_likelihood=lambda ls: float("-inf")
                  ) -> Optional[float]:
        """Log joint score of a point under its label; None if the label is
        unseen. Unseen feature values score via default_likelihood(existing
        log-likelihoods of that position) (:117-139)."""
        label = point.label
        if label not in self.priors:
            return None
        lls = self.likelihoods[label]
        if len(point.features) != len(lls):
            raise ValueError("feature arity mismatch")
        s = self.priors[label]
        for pos, v in enumerate(point.features):
            table = lls[pos]
            s += table.get(v, default_likelihood(list(table.values())))
        return s

    def predict(self, features: Sequence[str]) -> str:
        """argmax label (:141-153)."""
        best, best_s = None, float("-inf")
        for label in self.priors:
            s = self.log_score(LabeledPoint(label, features))
            if s is not None and s > best_s:
                best, best_s = label, s
        return best


class CategoricalNaiveBayes:
    """train (CategoricalNaiveBayes.scala:29-79)."""

    @staticmethod
    def train(points: Sequence[LabeledPoint]) -> CategoricalNaiveBayesModel:
        if not points:
            raise ValueError("no training points")
        arity = len(points[0].features)
        label_count: Dict[str, int] = defaultdict(int)
        feat_count: Dict[str, List[Dict[str, int]]] = {}
        for p in points:
            label_count[p.label] += 1
            fc = feat_count.setdefault(
                p.label, [defaultdict(int) for _ in range(arity)])
            for pos, v in enumerate(p.features):
                fc[pos][v] += 1
        n = len(points)
        priors = {lb: math.log(c / n) for lb, c in label_count.items()}
        likelihoods = {
            lb: [{v: math.log(c / label_count[lb])
                  for v, c in table.items()}
                 for table in feat_count[lb]]
            for lb in label_count
        }
        return CategoricalNaiveBayesModel(priors, likelihoods)


# ------------------------------------------------------- binary vectorizer

@dataclass
class BinaryVectorizer:
    """(field, value) pairs → one-hot vector (BinaryVectorizer.scala:26-63)."""
    property_map: Dict[Tuple[str, str], int]

    @staticmethod
    def fit(maps: Sequence[Dict[str, str]],
            properties: Sequence[str]) -> "BinaryVectorizer":
        seen = sorted({(f, m[f]) for m in maps for f in properties
                       if f in m})
        return BinaryVectorizer({kv: i for i, kv in enumerate(seen)})

    @property
    def num_features(self) -> int:
        return len(self.property_map)

    def to_vector(self, m: Dict[str, str]) -> torch.Tensor:
        v = torch.zeros(self.num_features)
        for kv, i in self.property_map.items():
            if m.get(kv[0]) == kv[1]:
                v[i] = 1.0
        return v


# ------------------------------------------------------- markov chain

@dataclass
class MarkovChainModel:
    """Row-normalized sparse transition matrix with per-row top-N
    (MarkovChain.scala:28-87)."""
    transition: Dict[int, List[Tuple[int, float]]]  # row → [(col, prob)]
    n: int
    top_n: int

    def transition_probs(self, state: int) -> List[Tuple[int, float]]:
        return self.transition.get(state, [])


class MarkovChain:
    @staticmethod
    def train(pairs: Sequence[Tuple[int, int]], n: int,
              top_n: int = 10) -> MarkovChainModel:
        counts: Dict[int, Dict[int, int]] = defaultdict(
            lambda: defaultdict(int))
        for a, b in pairs:
            counts[a][b] += 1
        transition = {}
        for a, row in counts.items():
            total = sum(row.values())
            probs = sorted(((b, c / total) for b, c in row.items()),
                           key=lambda t: -t[1])[:top_n]
            transition[a] = probs
        return MarkovChainModel(transition, n, top_n)


# ------------------------------------------------------- cross validation

def k_fold(data: Sequence, k: int):
    """Yield (training, testing) per fold — element i goes to test fold
    i % k (CrossValidation.scala:33-67 zipWithUniqueId % k semantics)."""
    for fold in range(k):
        train = [x for i, x in enumerate(data) if i % k != fold]
        test = [x for i, x in enumerate(data) if i % k == fold]
        yield train, test
