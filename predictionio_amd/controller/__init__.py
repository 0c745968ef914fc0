"""DASE controller API (reference: core/.../controller/)."""

from predictionio_amd.controller.base import (
    Algorithm, AverageServing, DataSource, EmptyParams, FirstServing,
    IdentityPreparator, Params, PersistentModel, Preparator, SanityCheck,
    Serving, resolve_class,
)
from predictionio_amd.controller.engine import (
    Engine, EngineFactory, EngineParams, PersistentModelManifest, get_engine,
)
from predictionio_amd.controller.metrics import (
    AverageMetric, EngineParamsGenerator, Evaluation, Metric, MetricEvaluator,
    MetricEvaluatorResult, OptionAverageMetric, StdevMetric, SumMetric,
    ZeroMetric,
)

__all__ = [
    "Algorithm", "AverageServing", "DataSource", "EmptyParams",
    "FirstServing", "IdentityPreparator", "Params", "PersistentModel",
    "Preparator", "SanityCheck", "Serving", "resolve_class",
    "Engine", "EngineFactory", "EngineParams", "PersistentModelManifest",
    "get_engine",
    "AverageMetric", "EngineParamsGenerator", "Evaluation", "Metric",
    "MetricEvaluator", "MetricEvaluatorResult", "OptionAverageMetric",
    "StdevMetric", "SumMetric", "ZeroMetric",
]
