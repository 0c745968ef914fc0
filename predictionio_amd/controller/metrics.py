"""Evaluation metrics + MetricEvaluator.

Parity with the reference metric stack:
- Metric.calculate (core/.../controller/Metric.scala:39-58)
- AverageMetric (:99-112), OptionAverageMetric (:124), StdevMetric,
  SumMetric, ZeroMetric (Metric.scala to :269)
- MetricEvaluator: ranks EngineParams candidates by the primary metric,
  writes best.json (MetricEvaluator.scala:185-263)
- Evaluation + EngineParamsGenerator
  (Evaluation.scala:34, EngineParamsGenerator.scala:29-46)
"""

from __future__ import annotations

import json
import math
from dataclasses import dataclass, field
from typing import Any, Callable, List, Optional, Sequence, Tuple


class Metric:
    """calculate(eval_data) → float, where eval_data is the Engine.eval
    output: [(eval_info, [(query, prediction, actual), ...]), ...]."""

    #: larger-is-better by default (Metric.scala `compare`)
    higher_is_better = True

    def header(self) -> str:
        return type(self).__name__

    def calculate(self, eval_data) -> float:
        raise NotImplementedError

    def compare(self, a: float, b: float) -> int:
        if a == b:
            return 0
        better = a > b if self.higher_is_better else a < b
        return 1 if better else -1


class AverageMetric(Metric):
    """Mean of a per-(q, p, a) score over all folds (Metric.scala:99-112)."""

    def calculate_one(self, eval_info, query, prediction, actual) -> float:
        raise NotImplementedError

    def calculate(self, eval_data) -> float:
        total, n = 0.0, 0
        for eval_info, qpa in eval_data:
            for q, p, a in qpa:
                total += self.calculate_one(eval_info, q, p, a)
                n += 1
        return total / n if n else float("nan")


class OptionAverageMetric(AverageMetric):
    """Mean over non-None scores only (Metric.scala:124)."""

    def calculate(self, eval_data) -> float:
        total, n = 0.0, 0
        for eval_info, qpa in eval_data:
            for q, p, a in qpa:
                s = self.calculate_one(eval_info, q, p, a)
                if s is not None:
                    total += s
                    n += 1
        return total / n if n else float("nan")


class StdevMetric(Metric):
    """Population stdev of per-tuple scores (Metric.scala StdevMetric)."""

    def calculate_one(self, eval_info, query, prediction, actual) -> float:
        raise NotImplementedError

    def calculate(self, eval_data) -> float:
        xs = [self.calculate_one(ei, q, p, a)
              for ei, qpa in eval_data for q, p, a in qpa]
        if not xs:
            return float("nan")
        mean = sum(xs) / len(xs)
        return math.sqrt(sum((x - mean) ** 2 for x in xs) / len(xs))


class SumMetric(Metric):
    def calculate_one(self, eval_info, query, prediction, actual) -> float:
        raise NotImplementedError

    def calculate(self, eval_data) -> float:
        return sum(self.calculate_one(ei, q, p, a)
                   for ei, qpa in eval_data for q, p, a in qpa)


class ZeroMetric(Metric):
    """Always 0 (Metric.scala ZeroMetric) — placeholder for dry runs."""

    def calculate(self, eval_data) -> float:
        return 0.0


@dataclass
class MetricEvaluatorResult:
    best_score: float
    best_engine_params: Any
    best_idx: int
    metric_header: str
    other_metric_headers: List[str]
    engine_params_scores: List[Tuple[Any, float, List[float]]]

    def summary(self) -> str:
        """One-line-per-candidate report (MetricEvaluator.scala logs +
        MetricEvaluatorResult.toOneLiner)."""
        lines = [f"Metric: {self.metric_header}"]
        for i, (ep, s, others) in enumerate(self.engine_params_scores):
            mark = " (best)" if i == self.best_idx else ""
            extra = "".join(f" {h}={v:.6f}" for h, v in
                            zip(self.other_metric_headers, others))
            lines.append(f"  [{i}] score={s:.6f}{extra}{mark}")
        lines.append(f"Best score: {self.best_score:.6f} "
                     f"(candidate {self.best_idx})")
        return "\n".join(lines)

    def to_json(self) -> dict:
        from dataclasses import asdict, is_dataclass
        ep = self.best_engine_params
        return {
            "bestScore": self.best_score,
            "bestIdx": self.best_idx,
            "metric": self.metric_header,
            "bestEngineParams": asdict(ep) if is_dataclass(ep) else str(ep),
            "scores": [s for _, s, _ in self.engine_params_scores],
        }


class MetricEvaluator:
    """Ranks candidate EngineParams by the primary metric and optionally
    writes best.json (MetricEvaluator.scala:185-263)."""

    def __init__(self, metric: Metric,
                 other_metrics: Optional[Sequence[Metric]] = None,
                 output_path: Optional[str] = None):
        self.metric = metric
        self.other_metrics = list(other_metrics or [])
        self.output_path = output_path

    def evaluate_base(self, engine, batch_eval_results) -> MetricEvaluatorResult:
        scored = []
        for ep, eval_data in batch_eval_results:
            primary = self.metric.calculate(eval_data)
            others = [m.calculate(eval_data) for m in self.other_metrics]
            scored.append((ep, primary, others))
        best_idx = 0
        for i, (_, s, _) in enumerate(scored):
            if self.metric.compare(s, scored[best_idx][1]) > 0:
                best_idx = i
        result = MetricEvaluatorResult(
            best_score=scored[best_idx][1],
            best_engine_params=scored[best_idx][0],
            best_idx=best_idx,
            metric_header=self.metric.header(),
            other_metric_headers=[m.header() for m in self.other_metrics],
            engine_params_scores=scored)
        if self.output_path:
            self._write_best_json(result)
        return result

    def _write_best_json(self, result: MetricEvaluatorResult) -> None:
        from dataclasses import asdict, is_dataclass
        ep = result.best_engine_params
        with open(self.output_path, "w") as f:
            json.dump({
                "bestScore": result.best_score,
                "metric": result.metric_header,
                "bestEngineParams": asdict(ep) if is_dataclass(ep) else str(ep),
            }, f, indent=2, default=str)


class EngineParamsGenerator:
    """Hyperparameter-candidate list (EngineParamsGenerator.scala:29-46)."""

    engine_params_list: List[Any] = []


class Evaluation:
    """An evaluation run = engine + evaluator (+ generator)
    (Evaluation.scala:34). Subclass and set the class attributes, or
    construct directly."""

    def __init__(self, engine=None, metric: Optional[Metric] = None,
                 evaluator: Optional[MetricEvaluator] = None,
                 engine_params_generator: Optional[EngineParamsGenerator] = None):
        self.engine = engine
        self.evaluator = evaluator or (MetricEvaluator(metric) if metric else None)
        self.engine_params_generator = engine_params_generator

    def run(self, engine_params_list: Optional[List[Any]] = None
            ) -> MetricEvaluatorResult:
        eps = engine_params_list
        if eps is None and self.engine_params_generator is not None:
            eps = self.engine_params_generator.engine_params_list
        if not eps:
            raise ValueError("No EngineParams candidates to evaluate")
        batch = self.engine.batch_eval(eps)
        return self.evaluator.evaluate_base(self.engine, batch)
