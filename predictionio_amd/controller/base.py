"""DASE controller API — the engine-developer-facing SDK.

Parity with the reference controller layer (core/.../controller/):
- Params (Params.scala:26-34), EngineParams (EngineParams.scala:35-55)
- DataSource: PDataSource/LDataSource readTraining/readEval
- Preparator: PPreparator/LPreparator/IdentityPreparator
- Algorithm: PAlgorithm/P2LAlgorithm/LAlgorithm train/predict/batchPredict +
  the three model-persistence modes (BaseAlgorithm.makePersistentModel,
  BaseAlgorithm.scala:95-115; PersistentModel.scala:67-102)
- Serving: LServing supplement/serve, LFirstServing, LAverageServing

Design shift for MI355X: the reference's P vs P2L vs L split encodes *where
the Spark RDD lives*. Here there is no executor fleet — a model is either
device-resident (torch tensors on one or more MI355X GPUs) or host-resident —
so a single Algorithm base class covers all three, and `persist_model`
chooses the persistence mode exactly as the reference does:
  return a bytes-serializable model  → auto-persisted (P2L/L behavior)
  implement PersistentModel          → manual save/load (PAlgorithm behavior)
  return None                        → not persisted, retrained at deploy
"""

from __future__ import annotations

import importlib
from typing import Any, Dict, List, Optional, Sequence, Tuple


class Params(dict):
    """Component parameters — a dict with attribute access (Params.scala)."""

    def __getattr__(self, k):
        try:
            return self[k]
        except KeyError as e:
            raise AttributeError(k) from e

    def __setattr__(self, k, v):
        self[k] = v


EmptyParams = Params


class SanityCheck:
    """Opt-in data sanity hook (core/.../core/SanityCheck.scala:25-33);
    called on training/prepared data and models unless --skip-sanity-check."""

    def sanity_check(self) -> None:
        pass


class DataSource:
    """Reads training + evaluation data (PDataSource.scala / LDataSource)."""

    def __init__(self, params: Optional[Params] = None):
        self.params = params or Params()

    def read_training(self) -> Any:
        raise NotImplementedError

    def read_eval(self) -> List[Tuple[Any, Any, List[Tuple[Any, Any]]]]:
        """Returns [(training_data, eval_info, [(query, actual), ...]), ...]
        — one tuple per evaluation fold (PDataSource.readEval)."""
        raise NotImplementedError


class Preparator:
    """Transforms TrainingData → PreparedData (PPreparator/LPreparator)."""

    def __init__(self, params: Optional[Params] = None):
        self.params = params or Params()

    def prepare(self, training_data: Any) -> Any:
        raise NotImplementedError


class IdentityPreparator(Preparator):
    """Pass-through preparator (IdentityPreparator.scala:32-56)."""

    def prepare(self, training_data: Any) -> Any:
        return training_data


class PersistentModel:
    """Custom-persistence contract (PersistentModel.scala:67-102): the model
    saves itself (e.g. factor tensors as .pt shards) and a manifest naming
    its class is stored in the model repository instead of a blob."""

    def save(self, instance_id: str, params: Params) -> bool:
        raise NotImplementedError

    @classmethod
    def load(cls, instance_id: str, params: Params) -> "PersistentModel":
        raise NotImplementedError


class Algorithm:
    """Train + predict (BaseAlgorithm.scala:58-126 and the P/P2L/L variants)."""

    def __init__(self, params: Optional[Params] = None):
        self.params = params or Params()

    def train(self, prepared_data: Any) -> Any:
        raise NotImplementedError

    def predict(self, model: Any, query: Any) -> Any:
        raise NotImplementedError

    def batch_predict(self, model: Any, queries: Sequence[Tuple[int, Any]]
                      ) -> List[Tuple[int, Any]]:
        """Default: map predict over (index, query) pairs
        (P2LAlgorithm.batchPredict, P2LAlgorithm.scala:69-71). Device-batched
        algorithms override this with a single fused kernel launch."""
        return [(i, self.predict(model, q)) for i, q in queries]

    def persist_model(self, model: Any) -> Any:
        """Choose the persistence mode (BaseAlgorithm.makePersistentModel):
        default auto-persists the model object via pickle. Return None to
        skip persistence (retrain at deploy); models implementing
        PersistentModel are saved via their own save()."""
        return model


class Serving:
    """Combines per-algorithm predictions (LServing.scala:30-55)."""

    def __init__(self, params: Optional[Params] = None):
        self.params = params or Params()

    def supplement(self, query: Any) -> Any:
        """Pre-predict query hook (LServing.supplementBase)."""
        return query

    def serve(self, query: Any, predictions: List[Any]) -> Any:
        raise NotImplementedError


class FirstServing(Serving):
    """Serve the first algorithm's prediction (LFirstServing.scala)."""

    def serve(self, query: Any, predictions: List[Any]) -> Any:
        return predictions[0]


class AverageServing(Serving):
    """Average numeric predictions (LAverageServing.scala)."""

    def serve(self, query: Any, predictions: List[Any]) -> Any:
        return sum(predictions) / len(predictions)


def resolve_class(path: str):
    """Import `pkg.module.Name` → class (the reference reflects engine
    factories by class name, WorkflowUtils.getEngine:53-69)."""
    module, _, name = path.rpartition(".")
    if not module:
        raise ValueError(f"Not a fully-qualified class name: {path}")
    return getattr(importlib.import_module(module), name)
