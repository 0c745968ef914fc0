"""End-to-end integration scenario — the quickstart lifecycle.

Mirrors tests/pio_tests/scenarios/quickstart_test.py from the reference:
app new → import events → train (via CLI) → deploy → HTTP queries
asserting itemScores — all in-process (CliRunner + TestClient) on the
recommendation template with synthetic MovieLens-like events.
"""

import json
import random

import pytest
from click.testing import CliRunner
from fastapi.testclient import TestClient

from predictionio_amd.cli.main import cli


@pytest.fixture()
def env(mem_storage, tmp_path):
    return {"storage": mem_storage, "tmp": tmp_path,
            "runner": CliRunner()}


def _seed_events_file(path, n_users=25, n_items=15):
    rng = random.Random(42)
    evs = []
    for u in range(n_users):
        liked = [i for i in range(n_items) if i % 2 == u % 2]
        for i in rng.sample(liked, 5):
            evs.append({"event": "rate", "entityType": "user",
                        "entityId": f"u{u}", "targetEntityType": "item",
                        "targetEntityId": f"i{i}",
                        "properties": {"rating": rng.uniform(3.5, 5.0)}})
    path.write_text("\n".join(json.dumps(e) for e in evs))
    return len(evs)


class TestQuickstart:
    def test_full_lifecycle(self, env):
        runner, tmp = env["runner"], env["tmp"]

        # 1. app new
        r = runner.invoke(cli, ["app", "new", "QuickApp",
                                "--access-key", "QKEY"])
        assert r.exit_code == 0
        from predictionio_amd.data import storage
        app_id = storage.get_meta_data_apps().get_by_name("QuickApp").id

        # 2. event ingest through the REST event server (SDK-style)
        from predictionio_amd.server.eventserver import create_app
        es = TestClient(create_app())
        n = 0
        f = tmp / "events.json"
        _seed_events_file(f)
        batch = []
        for line in f.read_text().splitlines():
            batch.append(json.loads(line))
            if len(batch) == 50:
                rr = es.post("/batch/events.json?accessKey=QKEY",
                             json=batch)
                assert rr.status_code == 200
                n += sum(1 for x in rr.json() if x["status"] == 201)
                batch = []
        if batch:
            rr = es.post("/batch/events.json?accessKey=QKEY", json=batch)
            n += sum(1 for x in rr.json() if x["status"] == 201)
        assert n == 125

        # 3. engine dir from the built-in template + train via CLI
        engine_dir = tmp / "engine"
        r = runner.invoke(cli, ["template", "get", "recommendation",
                                str(engine_dir)])
        assert r.exit_code == 0
        variant = json.loads((engine_dir / "engine.json").read_text())
        variant["datasource"]["params"]["appName"] = "QuickApp"
        variant["algorithms"][0]["params"].update(
            {"rank": 8, "numIterations": 5})
        (engine_dir / "engine.json").write_text(json.dumps(variant))
        r = runner.invoke(cli, ["train", "--engine-dir", str(engine_dir)])
        assert r.exit_code == 0, r.output

        # 4. deploy + query
        from predictionio_amd.server.queryserver import (
            ServerConfig, create_app as qs_app,
        )
        qs = TestClient(qs_app(ServerConfig(
            engine_factory=variant["engineFactory"],
            engine_variant=variant["id"])))
        r = qs.post("/queries.json", json={"user": "u1", "num": 4})
        assert r.status_code == 200
        scores = r.json()["itemScores"]
        assert len(scores) == 4
        assert all("item" in s and "score" in s for s in scores)
        # 5. status page shows the request
        assert qs.get("/status.json").json()["requestCount"] == 1

    def test_classification_lifecycle(self, env):
        """BASELINE config 1: classification on CPU via the event server."""
        runner, tmp = env["runner"], env["tmp"]
        runner.invoke(cli, ["app", "new", "ClassApp",
                            "--access-key", "CKEY"])
        from predictionio_amd.server.eventserver import create_app
        es = TestClient(create_app())
        rng = random.Random(1)
        for u in range(40):
            plan = u % 2
            r = es.post("/events.json?accessKey=CKEY", json={
                "event": "$set", "entityType": "user",
                "entityId": f"u{u}",
                "properties": {
                    "plan": plan,
                    "attr0": rng.uniform(8, 10) if plan == 0
                    else rng.uniform(0, 2),
                    "attr1": 5.0,
                    "attr2": rng.uniform(0, 2) if plan == 0
                    else rng.uniform(8, 10)}})
            assert r.status_code == 201
        engine_dir = tmp / "cls"
        runner.invoke(cli, ["template", "get", "classification",
                            str(engine_dir)])
        variant = json.loads((engine_dir / "engine.json").read_text())
        variant["datasource"]["params"]["appName"] = "ClassApp"
        (engine_dir / "engine.json").write_text(json.dumps(variant))
        r = runner.invoke(cli, ["train", "--engine-dir", str(engine_dir)])
        assert r.exit_code == 0, r.output
        from predictionio_amd.server.queryserver import (
            ServerConfig, create_app as qs_app,
        )
        qs = TestClient(qs_app(ServerConfig(
            engine_factory=variant["engineFactory"])))
        assert qs.post("/queries.json",
                       json={"attr0": 9, "attr1": 5,
                             "attr2": 1}).json() == {"label": 0.0}
        assert qs.post("/queries.json",
                       json={"attr0": 1, "attr1": 5,
                             "attr2": 9}).json() == {"label": 1.0}


class TestFsspecModels:
    def test_model_blob_roundtrip(self, tmp_path, monkeypatch):
        from predictionio_amd.data import storage
        storage.reset()
        monkeypatch.setenv("PIO_FS_BASEDIR", str(tmp_path))
        monkeypatch.setenv("PIO_STORAGE_SOURCES_META_TYPE", "sqlite")
        monkeypatch.setenv("PIO_STORAGE_SOURCES_META_PATH",
                           str(tmp_path / "m.db"))
        monkeypatch.setenv("PIO_STORAGE_SOURCES_BLOB_TYPE", "fsspec")
        monkeypatch.setenv("PIO_STORAGE_SOURCES_BLOB_PATH",
                           f"file://{tmp_path}/blobs")
        for repo, src in (("METADATA", "META"), ("EVENTDATA", "META"),
                          ("MODELDATA", "BLOB")):
            monkeypatch.setenv(f"PIO_STORAGE_REPOSITORIES_{repo}_SOURCE",
                               src)
            monkeypatch.setenv(f"PIO_STORAGE_REPOSITORIES_{repo}_NAME",
                               repo.lower())
        from predictionio_amd.data.storage.base import Model
        models = storage.get_model_data_models()
        models.insert(Model("abc", b"\x00\x01binary"))
        got = models.get("abc")
        assert got is not None and got.models == b"\x00\x01binary"
        assert models.delete("abc") is True
        assert models.get("abc") is None
        storage.reset()
