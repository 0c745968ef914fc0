"""Evaluation workflow: runs an Evaluation and records an
EvaluationInstance.

Parity with the reference evaluation path:
- EvaluationWorkflow.runEvaluation = engine.batchEval + evaluator
  (core/.../workflow/EvaluationWorkflow.scala:31-45)
- Workflow.runEvaluation wires Evaluation + EngineParamsGenerator and
  persists the EvaluationInstance (Workflow.scala:82-138,
  CoreWorkflow.runEvaluation :104-164)
- `pio eval Evaluation [Generator]` resolves the classes by name
  (CreateWorkflow.main :257-276)
"""

from __future__ import annotations

import json
import logging
from typing import Any, List, Optional, Tuple

from predictionio_amd.controller.base import resolve_class
from predictionio_amd.controller.metrics import (
    EngineParamsGenerator, Evaluation, MetricEvaluatorResult,
)
from predictionio_amd.data import storage
from predictionio_amd.data.events import utcnow
from predictionio_amd.data.storage.base import EvaluationInstance

log = logging.getLogger(__name__)


def run_evaluation(evaluation: Evaluation,
                   engine_params_list: Optional[List[Any]] = None,
                   evaluation_class: str = "",
                   generator_class: str = "",
                   batch: str = "") -> Tuple[str, MetricEvaluatorResult]:
    """Run + persist. Returns (evaluation instance id, result)."""
    instances = storage.get_meta_data_evaluation_instances()
    inst = EvaluationInstance(
        id="", status="INIT", start_time=utcnow(), end_time=utcnow(),
        evaluation_class=evaluation_class,
        engine_params_generator_class=generator_class, batch=batch)
    iid = instances.insert(inst)
    inst.id = iid
    inst.status = "EVALUATING"
    instances.update(inst)
    try:
        result = evaluation.run(engine_params_list)
    except Exception:
        inst.status = "FAILED"
        inst.end_time = utcnow()
        instances.update(inst)
        raise
    inst.status = "EVALCOMPLETED"
    inst.end_time = utcnow()
    inst.evaluator_results = result.summary()
    inst.evaluator_results_json = json.dumps(result.to_json())
    inst.evaluator_results_html = (
        "<pre>" + result.summary() + "</pre>")
    instances.update(inst)
    log.info("Evaluation completed: instance %s", iid)
    return iid, result


def _instantiate(path: str):
    obj = resolve_class(path)
    return obj() if isinstance(obj, type) else obj


def run_evaluation_classes(evaluation_class: str,
                           generator_class: Optional[str] = None,
                           batch: str = ""
                           ) -> Tuple[str, MetricEvaluatorResult]:
    """`pio eval pkg.mod.MyEvaluation [pkg.mod.MyGenerator]`."""
    evaluation = _instantiate(evaluation_class)
    if not isinstance(evaluation, Evaluation):
        raise TypeError(f"{evaluation_class} is not an Evaluation")
    eps = None
    if generator_class:
        gen = _instantiate(generator_class)
        if not isinstance(gen, EngineParamsGenerator):
            raise TypeError(f"{generator_class} is not an "
                            "EngineParamsGenerator")
        eps = gen.engine_params_list
    return run_evaluation(evaluation, eps,
                          evaluation_class=evaluation_class,
                          generator_class=generator_class or "",
                          batch=batch)
