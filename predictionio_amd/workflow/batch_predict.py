"""Batch predict: query-per-line JSON file → prediction-per-line file.

Parity with core/.../workflow/BatchPredict.scala:145-234: load the latest
completed instance's models, then for each input line run
supplement → predict per algorithm → serve → JSON line. The reference
repartitions a Spark RDD (:192-195); here device-batched algorithms
override Algorithm.batch_predict with one fused kernel launch over the
whole query set, and the per-line path is the fallback.
"""

from __future__ import annotations

import json
import logging
from typing import Dict, Optional

log = logging.getLogger(__name__)


def run_batch_predict(variant: Dict, input_path: str, output_path: str,
                      engine_instance_id: Optional[str] = None) -> int:
    from predictionio_amd.server.queryserver import ServerConfig, _load_state

    cfg = ServerConfig(engine_factory=variant["engineFactory"],
                       engine_variant=variant.get("id", "default"),
                       engine_instance_id=engine_instance_id)
    st = _load_state(cfg)

    with open(input_path) as f:
        queries = [json.loads(line) for line in f if line.strip()]

    # typed queries when the algorithm declares a parser (the server
    # route does the same; JsonExtractor semantics)
    conv = getattr(st.algorithms[0], "query_from_json", None)
    typed = [conv(q) if conv else q for q in queries]
    # supplement each query, then let each algorithm batch-predict (device
    # algorithms fuse this into one kernel launch)
    supplemented = [st.serving.supplement(q) for q in typed]
    indexed = list(enumerate(supplemented))
    per_algo = []
    for algo, model in zip(st.algorithms, st.models):
        per_algo.append(dict(algo.batch_predict(model, indexed)))

    n = 0
    with open(output_path, "w") as out:
        for i, q in enumerate(queries):
            preds = [pa[i] for pa in per_algo]
            result = st.serving.serve(typed[i], preds)
            rj = result.to_json() if hasattr(result, "to_json") else result
            out.write(json.dumps({"query": q, "prediction": rj}) + "\n")
            n += 1
    log.info("Batch predict: %d predictions -> %s", n, output_path)
    return n
