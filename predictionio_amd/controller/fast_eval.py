"""FastEvalEngine — stage-result caching for hyperparameter search.

Parity with core/.../controller/FastEvalEngine.scala:46-346: during
`batch_eval` over many EngineParams candidates, results of each DASE stage
are cached keyed by the parameter PREFIX that determines them
(DataSourcePrefix → eval folds; PreparatorPrefix → prepared data;
AlgorithmsPrefix → trained models + batch predictions; ServingPrefix →
served results), so candidates sharing upstream params reuse upstream
work. The reference trades the exact full-workflow semantics for speed
(FastEvalEngine.scala header comment) — same here.
"""

from __future__ import annotations

import json
from typing import Any, Dict, List, Tuple

from predictionio_amd.controller.engine import Engine, EngineParams


def _key(*parts: Any) -> str:
    return json.dumps(parts, sort_keys=True, default=str)


class FastEvalEngine(Engine):
    """Drop-in Engine whose batch_eval memoizes per-stage results.

    Cache hit counters (`hits`/`misses` per stage) are exposed for tests —
    the reference's FastEvalEngineTest asserts exactly this."""

    def __init__(self, *args, **kw):
        super().__init__(*args, **kw)
        self._cache: Dict[str, Any] = {}
        self.hits: Dict[str, int] = {"datasource": 0, "preparator": 0,
                                     "algorithms": 0}
        self.misses: Dict[str, int] = {"datasource": 0, "preparator": 0,
                                       "algorithms": 0}

    def _cached(self, stage: str, key: str, compute):
        if key in self._cache:
            self.hits[stage] += 1
            return self._cache[key]
        self.misses[stage] += 1
        v = compute()
        self._cache[key] = v
        return v

    def _eval_folds(self, ep: EngineParams):
        k = _key("ds", ep.data_source_name, ep.data_source_params)
        return self._cached(
            "datasource", k,
            lambda: self._data_source(ep).read_eval())

    def _prepared(self, ep: EngineParams):
        k = _key("prep", ep.data_source_name, ep.data_source_params,
                 ep.preparator_name, ep.preparator_params)
        prep = self._preparator(ep)
        return self._cached(
            "preparator", k,
            lambda: [(prep.prepare(td), ei, qa)
                     for td, ei, qa in self._eval_folds(ep)])

    def _predictions(self, ep: EngineParams):
        k = _key("algo", ep.data_source_name, ep.data_source_params,
                 ep.preparator_name, ep.preparator_params,
                 ep.algorithms_params)

        def compute():
            out = []
            for pd, ei, qa in self._prepared(ep):
                algos = self._algorithms(ep)
                models = [a.train(pd) for a in algos]
                queries = [(i, q) for i, (q, _) in enumerate(qa)]
                per_algo = [dict(a.batch_predict(m, queries))
                            for a, m in zip(algos, models)]
                out.append((ei, qa, per_algo))
            return out

        return self._cached("algorithms", k, compute)

    def eval(self, engine_params: EngineParams):
        serving = self._serving(engine_params)
        result = []
        for ei, qa, per_algo in self._predictions(engine_params):
            qpa = []
            for i, (q, actual) in enumerate(qa):
                preds = [pa[i] for pa in per_algo]
                qpa.append((q, serving.serve(q, preds), actual))
            result.append((ei, qpa))
        return result

    def clear_cache(self) -> None:
        self._cache.clear()
