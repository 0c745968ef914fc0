"""Driver-contract test for bench.py: the JSON line the round driver
parses must keep its schema (keys, types, units) in both modes."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {
    "metric": str, "value": (int, float), "unit": str, "n_gpus": int,
    "steps": int, "warmup": int, "ms_per_step": (int, float),
    "higher_is_better": bool, "scaling": str, "dtype": str, "data": str,
    "config": dict,
}


def _run(mode):
    out = subprocess.run(
        [sys.executable, "bench.py", "--cpu-small", "--steps", "1",
         "--warmup", "0", "--mode", mode],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-800:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert lines, f"no JSON line in stdout: {out.stdout[-300:]}"
    return json.loads(lines[-1])

class TestBenchContract:
    def test_train_mode_json(self):
        j = _run("train")
        for k, t in REQUIRED.items():
            assert k in j, f"missing key {k}"
            assert isinstance(j[k], t), (k, type(j[k]))
        assert "vs_baseline" in j  # null allowed (no published baseline)
        assert j["metric"] == "als_ratings_per_sec"
        assert j["unit"] == "ratings/s"
        assert j["n_gpus"] == 1 and j["steps"] == 1
        assert j["scaling"] == "weak" and j["data"] == "synthetic"
        assert j["value"] > 0 and j["ms_per_step"] > 0
        # whole-job aggregate consistency: value = nnz / sec_per_iter
        cfg = j["config"]
        assert abs(j["value"] - cfg["global_batch"] /
                   (j["ms_per_step"] / 1e3)) / j["value"] < 1e-6
        assert cfg["parallelism"] == "dp1"

    def test_serve_mode_json(self):
        j = _run("serve")
        assert j["metric"] == "serving_queries_per_sec"
        assert j["unit"] == "queries/s"
        assert j["value"] > 0
        assert j["config"]["topk"] == 20
