"""Recommendation evaluation: Precision@K with k-fold data
(examples/scala-parallel-recommendation/*/src/main/scala/Evaluation.scala:
PrecisionAtK metric + RecommendationEvaluation)."""

from __future__ import annotations

from typing import List, Optional

from predictionio_amd.controller import (
    Evaluation, MetricEvaluator, OptionAverageMetric,
)


class PrecisionAtK(OptionAverageMetric):
    """Of the top-k recommended items, the fraction actually rated
    positively by the user; None when the user has no actuals (excluded
    from the mean — OptionAverageMetric semantics)."""

    def __init__(self, k: int = 10):
        self.k = k

    def header(self) -> str:
        return f"Precision@{self.k}"

    def calculate_one(self, eval_info, query, prediction, actual
                      ) -> Optional[float]:
        if not actual:
            return None
        positives = set(actual)
        top = [s.item for s in prediction.item_scores[:self.k]]
        if not top:
            return None
        return len([t for t in top if t in positives]) / float(len(top))


class RecommendationEvaluation(Evaluation):
    def __init__(self):
        from predictionio_amd.templates.recommendation import (
            RecommendationEngine,
        )
        super().__init__(engine=RecommendationEngine.apply(),
                         metric=PrecisionAtK(k=10))
