"""Engine: chains DataSource → Preparator → Algorithm(s) → Serving.

Parity with the reference Engine (core/.../controller/Engine.scala):
- class maps + EngineParams selection (Engine.scala:82-154)
- train (object Engine.train, Engine.scala:623-710) incl. sanity-check hooks
- eval (object Engine.eval, Engine.scala:728-817)
- prepareDeploy / model rehydration incl. retrain-if-not-persisted and
  PersistentModel manifests (Engine.scala:198-267)
- engine.json parsing (jValueToEngineParams, Engine.scala:355-418)
- EngineParams from a stored EngineInstance (engineInstanceToEngineParams,
  Engine.scala:420-490)
"""

from __future__ import annotations

import json
import pickle
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Sequence, Tuple, Type

from predictionio_amd.controller.base import (
    Algorithm, DataSource, FirstServing, Params, PersistentModel, Preparator,
    SanityCheck, Serving, resolve_class,
)


@dataclass
class EngineParams:
    """Named per-stage parameter bundle (EngineParams.scala:35-55)."""
    data_source_name: str = ""
    data_source_params: Params = field(default_factory=Params)
    preparator_name: str = ""
    preparator_params: Params = field(default_factory=Params)
    # list of (algorithm-name, params) — order defines prediction order
    algorithms_params: List[Tuple[str, Params]] = field(default_factory=list)
    serving_name: str = ""
    serving_params: Params = field(default_factory=Params)


@dataclass
class PersistentModelManifest:
    """Stored in place of a model blob when the algorithm's model persists
    itself (workflow/PersistentModelManifest.scala:21)."""
    class_name: str


class _NotPersisted:
    """Marker stored when persist_model() returned None — the model is
    retrained at deploy (Engine.prepareDeploy, Engine.scala:210-228)."""

    def __repr__(self):
        return "NotPersisted"


def _as_class_map(x) -> Dict[str, type]:
    if isinstance(x, dict):
        return x
    return {"": x}


class Engine:
    """An engine = class maps for the four DASE stages.

    `data_source_class` etc. accept either a single class (mapped under the
    default name "") or a dict of name→class for multi-variant engines."""

    def __init__(self,
                 data_source_class,
                 preparator_class,
                 algorithm_class,
                 serving_class):
        self.data_source_class_map = _as_class_map(data_source_class)
        self.preparator_class_map = _as_class_map(preparator_class)
        self.algorithm_class_map = _as_class_map(algorithm_class)
        self.serving_class_map = _as_class_map(serving_class)

    # ------------------------------------------------------------ factories

    def _data_source(self, ep: EngineParams) -> DataSource:
        cls = self.data_source_class_map[ep.data_source_name]
        return cls(ep.data_source_params)

    def _preparator(self, ep: EngineParams) -> Preparator:
        cls = self.preparator_class_map[ep.preparator_name]
        return cls(ep.preparator_params)

    def _algorithms(self, ep: EngineParams) -> List[Algorithm]:
        algos = []
        for name, params in (ep.algorithms_params or [("", Params())]):
            cls = self.algorithm_class_map[name]
            algos.append(cls(params))
        return algos

    def _serving(self, ep: EngineParams) -> Serving:
        cls = self.serving_class_map[ep.serving_name]
        return cls(ep.serving_params)

    # ------------------------------------------------------------ training

    def train(self, engine_params: EngineParams,
              skip_sanity_check: bool = False) -> List[Any]:
        """Read → (sanity) → prepare → (sanity) → train each algorithm →
        (sanity). Returns the in-memory models, one per algorithm
        (object Engine.train, Engine.scala:623-710)."""
        ds = self._data_source(engine_params)
        td = ds.read_training()
        if not skip_sanity_check and isinstance(td, SanityCheck):
            td.sanity_check()
        prep = self._preparator(engine_params)
        pd = prep.prepare(td)
        if not skip_sanity_check and isinstance(pd, SanityCheck):
            pd.sanity_check()
        models = []
        for algo in self._algorithms(engine_params):
            m = algo.train(pd)
            if not skip_sanity_check and isinstance(m, SanityCheck):
                m.sanity_check()
            models.append(m)
        return models

    def make_serializable_models(self, engine_params: EngineParams,
                                 instance_id: str,
                                 models: List[Any]) -> bytes:
        """Serialize models for the Models repository. Per algorithm
        (Engine.makeSerializableModels, Engine.scala:284-302):
          PersistentModel → model.save() + manifest;
          persist_model()→None → NotPersisted marker;
          otherwise → pickled object (reference: Kryo blob,
          CoreWorkflow.scala:76-81)."""
        algos = self._algorithms(engine_params)
        entries: List[Any] = []
        for algo, model in zip(algos, models):
            if isinstance(model, PersistentModel):
                ok = model.save(instance_id, algo.params)
                entries.append(
                    PersistentModelManifest(_class_path(type(model)))
                    if ok else _NotPersisted())
            else:
                p = algo.persist_model(model)
                entries.append(p if p is not None else _NotPersisted())
        return pickle.dumps(entries, protocol=pickle.HIGHEST_PROTOCOL)

    def prepare_deploy(self, engine_params: EngineParams,
                       instance_id: str, blob: Optional[bytes]) -> List[Any]:
        """Rehydrate models for serving (Engine.prepareDeploy,
        Engine.scala:198-267): manifest → PersistentModel.load;
        NotPersisted → retrain now; blob → use directly."""
        algos = self._algorithms(engine_params)
        entries = pickle.loads(blob) if blob else [_NotPersisted()] * len(algos)
        retrained: Optional[List[Any]] = None
        models: List[Any] = []
        for i, (algo, entry) in enumerate(zip(algos, entries)):
            if isinstance(entry, PersistentModelManifest):
                cls = resolve_class(entry.class_name)
                models.append(cls.load(instance_id, algo.params))
            elif isinstance(entry, _NotPersisted):
                if retrained is None:
                    retrained = self.train(engine_params,
                                           skip_sanity_check=True)
                models.append(retrained[i])
            else:
                models.append(entry)
        return models

    # ------------------------------------------------------------ evaluation

    def eval(self, engine_params: EngineParams
             ) -> List[Tuple[Any, List[Tuple[Any, Any, Any]]]]:
        """k-fold evaluation (object Engine.eval, Engine.scala:728-817):
        per fold — prepare, train all algorithms, batch-predict each, combine
        per-query predictions through Serving. Returns
        [(eval_info, [(query, prediction, actual), ...]), ...]."""
        ds = self._data_source(engine_params)
        folds = ds.read_eval()
        prep = self._preparator(engine_params)
        serving = self._serving(engine_params)
        out = []
        for td, eval_info, qa_list in folds:
            pd = prep.prepare(td)
            algos = self._algorithms(engine_params)
            models = [a.train(pd) for a in algos]
            queries = [(i, q) for i, (q, _) in enumerate(qa_list)]
            per_algo: List[Dict[int, Any]] = []
            for a, m in zip(algos, models):
                per_algo.append(dict(a.batch_predict(m, queries)))
            qpa = []
            for i, (q, actual) in enumerate(qa_list):
                preds = [pa[i] for pa in per_algo]
                qpa.append((q, serving.serve(q, preds), actual))
            out.append((eval_info, qpa))
        return out

    def batch_eval(self, engine_params_list: List[EngineParams]):
        """(engine_params, eval result) per candidate (BaseEngine.batchEval)."""
        return [(ep, self.eval(ep)) for ep in engine_params_list]

    # ------------------------------------------------------------ engine.json

    def json_to_engine_params(self, variant: Dict[str, Any]) -> EngineParams:
        """Parse an engine.json variant (jValueToEngineParams,
        Engine.scala:355-418): optional `datasource`/`preparator`/`serving`
        objects with `params` (and optional `name`), and an `algorithms`
        array of {"name": ..., "params": ...}."""
        def stage(key) -> Tuple[str, Params]:
            obj = variant.get(key) or {}
            params = obj.get("params", obj if key != "algorithms" else {})
            return obj.get("name", ""), Params(params or {})

        ds_name, ds_params = stage("datasource")
        p_name, p_params = stage("preparator")
        s_name, s_params = stage("serving")
        algo_list = []
        for a in variant.get("algorithms", []):
            algo_list.append((a.get("name", ""), Params(a.get("params", {}))))
        if not algo_list:
            algo_list = [("", Params())]
        return EngineParams(
            data_source_name=ds_name, data_source_params=ds_params,
            preparator_name=p_name, preparator_params=p_params,
            algorithms_params=algo_list,
            serving_name=s_name, serving_params=s_params)

    def engine_instance_to_engine_params(self, instance) -> EngineParams:
        """Rebuild EngineParams from a stored EngineInstance
        (engineInstanceToEngineParams, Engine.scala:420-490)."""
        ds = json.loads(instance.data_source_params or "{}")
        prep = json.loads(instance.preparator_params or "{}")
        algos = json.loads(instance.algorithms_params or "[]")
        serv = json.loads(instance.serving_params or "{}")
        return EngineParams(
            data_source_name=ds.get("name", ""),
            data_source_params=Params(ds.get("params", {})),
            preparator_name=prep.get("name", ""),
            preparator_params=Params(prep.get("params", {})),
            algorithms_params=[(a.get("name", ""), Params(a.get("params", {})))
                               for a in algos] or [("", Params())],
            serving_name=serv.get("name", ""),
            serving_params=Params(serv.get("params", {})))


class EngineFactory:
    """Engine factory contract: subclasses implement apply() → Engine
    (controller/EngineFactory in the reference)."""

    @classmethod
    def apply(cls) -> Engine:
        raise NotImplementedError


def _class_path(cls: type) -> str:
    return f"{cls.__module__}.{cls.__qualname__}"


def get_engine(factory_path: str) -> Engine:
    """Resolve an engineFactory string → Engine instance
    (WorkflowUtils.getEngine, WorkflowUtils.scala:53-69)."""
    obj = resolve_class(factory_path)
    if isinstance(obj, Engine):
        return obj
    if isinstance(obj, type) and issubclass(obj, EngineFactory):
        return obj.apply()
    if callable(obj):
        e = obj()
        if isinstance(e, Engine):
            return e
    raise TypeError(f"{factory_path} is not an Engine factory")
