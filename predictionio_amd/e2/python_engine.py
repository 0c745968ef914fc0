"""Generic model-wrapping engine (reference: e2/.../PythonEngine.scala:31-96
wraps a Spark ML PipelineModel with no-op DataSource/Preparator and a
predict that runs model.transform on a single-row frame).

Here the equivalent wraps ANY picklable predictor exposing
`fit(rows) -> fitted` and `predict(fitted, query) -> result` (or an
sklearn-style object with .fit/.predict): useful for serving arbitrary
Python models behind the standard train/deploy workflow without writing a
full template.
"""

from __future__ import annotations

from typing import Any, Callable, List, Optional

from predictionio_amd.controller import (
    Algorithm, DataSource, Engine, FirstServing, IdentityPreparator, Params,
)
from predictionio_amd.controller.base import resolve_class


class EventDataSource(DataSource):
    """Reads raw events of an app (params: appName, [eventNames])."""

    def read_training(self):
        from predictionio_amd.data import event_store
        return event_store.find(
            self.params["appName"],
            event_names=self.params.get("eventNames"))


class PythonAlgorithm(Algorithm):
    """Params: estimatorClass — dotted path to an object with
    fit(training_data) and predict(model, query); or an sklearn-style
    class instantiated with estimatorParams."""

    def _estimator(self):
        cls = resolve_class(self.params["estimatorClass"])
        kw = self.params.get("estimatorParams", {})
        return cls(**kw) if isinstance(cls, type) else cls

    def train(self, prepared_data):
        est = self._estimator()
        if hasattr(est, "fit"):
            fitted = est.fit(prepared_data)
            # sklearn returns self; keep the estimator as the model
            return fitted if fitted is not None else est
        raise TypeError("estimator must define fit()")

    def predict(self, model, query):
        return model.predict(query)


def python_engine() -> Engine:
    return Engine(EventDataSource, IdentityPreparator, PythonAlgorithm,
                  FirstServing)


class PythonEngineFactory:
    @classmethod
    def apply(cls) -> Engine:
        return python_engine()
