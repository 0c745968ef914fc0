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


class TestMultiProcessTrain:
    def test_pio_train_gpus2_cpu(self, mem_storage, tmp_path):
        """`pio train --gpus 2` spawns torchrun (gloo on CPU): both ranks
        read the shared sqlite store, shard the ALS solve, rank 0
        persists — then the model deploys and answers queries."""
        import json as js
        from click.testing import CliRunner
        from predictionio_amd.cli.main import cli
        runner = CliRunner()
        r = runner.invoke(cli, ["app", "new", "DistApp",
                                "--access-key", "DKEY"])
        assert r.exit_code == 0
        from predictionio_amd.data import storage
        app_id = storage.get_meta_data_apps().get_by_name("DistApp").id
        le = storage.get_l_events()
        import random
        from datetime import datetime, timezone
        from predictionio_amd.data.events import DataMap, Event
        rng = random.Random(3)
        for u in range(20):
            liked = [i for i in range(12) if i % 2 == u % 2]
            for i in rng.sample(liked, 4):
                le.insert(Event(
                    event="rate", entity_type="user", entity_id=f"u{u}",
                    target_entity_type="item", target_entity_id=f"i{i}",
                    properties=DataMap({"rating": rng.uniform(3.5, 5.0)}),
                    event_time=datetime.now(timezone.utc)), app_id)
        engine_dir = tmp_path / "eng"
        r = runner.invoke(cli, ["template", "get", "recommendation",
                                str(engine_dir)])
        assert r.exit_code == 0
        variant = js.loads((engine_dir / "engine.json").read_text())
        variant["datasource"]["params"]["appName"] = "DistApp"
        variant["algorithms"][0]["params"].update(
            {"rank": 8, "numIterations": 3})
        (engine_dir / "engine.json").write_text(js.dumps(variant))
        r = runner.invoke(cli, ["train", "--engine-dir", str(engine_dir),
                                "--gpus", "2"])
        assert r.exit_code == 0, r.output
        # the COMPLETED instance was written by rank 0 of the torchrun job
        insts = storage.get_meta_data_engine_instances()
        done = [i for i in insts.get_all() if i.status == "COMPLETED"]
        assert done, "no COMPLETED engine instance from distributed train"
        from predictionio_amd.server.queryserver import (
            ServerConfig, create_app as qs_app,
        )
        qs = TestClient(qs_app(ServerConfig(
            engine_factory=variant["engineFactory"],
            engine_variant=variant["id"])))
        resp = qs.post("/queries.json", json={"user": "u1", "num": 3})
        assert resp.status_code == 200
        assert len(resp.json()["itemScores"]) == 3


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


class TestServerProcesses:
    """Launch the real `pio eventserver` / `pio deploy` commands as
    processes (the reference integration harness drives the CLI the same
    way, tests/pio_tests/utils.py AppEngine.deploy)."""

    def _spawn(self, args, env):
        import subprocess, sys, os, socket, time
        root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        env = dict(env)
        env["PYTHONPATH"] = root + os.pathsep + env.get("PYTHONPATH", "")
        proc = subprocess.Popen(
            [sys.executable, "-m", "predictionio_amd.cli.main", *args],
            env=env, cwd=root, stdout=subprocess.DEVNULL,
            stderr=subprocess.PIPE)
        return proc

    def _wait(self, proc, port, timeout=25):
        import socket, time
        t0 = time.time()
        while time.time() - t0 < timeout:
            if proc.poll() is not None:
                raise RuntimeError(proc.stderr.read().decode()[:400])
            try:
                with socket.create_connection(("127.0.0.1", port),
                                              timeout=0.3):
                    return
            except OSError:
                time.sleep(0.15)
        proc.terminate()
        raise RuntimeError("no listen")

    def test_pio_eventserver_and_deploy(self, mem_storage, tmp_path):
        import json as js
        import os
        import socket
        import urllib.request
        from predictionio_amd.data.storage.base import AccessKey, App
        app_id = mem_storage.get_meta_data_apps().insert(App(0, "ProcApp"))
        mem_storage.get_meta_data_access_keys().insert(
            AccessKey(key="PKEY", appid=app_id, events=[]))
        mem_storage.get_l_events().init(app_id)

        def free_port():
            s = socket.socket(); s.bind(("127.0.0.1", 0))
            p = s.getsockname()[1]; s.close(); return p

        ep_port = free_port()
        es = self._spawn(["eventserver", "--ip", "127.0.0.1",
                          "--port", str(ep_port)], os.environ)
        try:
            self._wait(es, ep_port)
            body = js.dumps({"event": "rate", "entityType": "user",
                             "entityId": "u1", "targetEntityType": "item",
                             "targetEntityId": "i1",
                             "properties": {"rating": 5}}).encode()
            req = urllib.request.Request(
                f"http://127.0.0.1:{ep_port}/events.json?accessKey=PKEY",
                data=body, headers={"Content-Type": "application/json"})
            assert js.loads(urllib.request.urlopen(req, timeout=5).read()
                            )["eventId"]
        finally:
            es.terminate(); es.wait(timeout=10)

        # train a JSON engine then `pio deploy` it as a process
        from predictionio_amd.workflow import train as train_wf
        variant = {"id": "proc",
                   "engineFactory": "tests.fake_engine.JsonEngineFactory",
                   "datasource": {"params": {"n": 4}},
                   "algorithms": [{"name": "", "params": {}}]}
        train_wf.run_train_from_variant(variant)
        vpath = tmp_path / "engine.json"
        vpath.write_text(js.dumps(variant))
        qs_port = free_port()
        qs = self._spawn(["deploy", "--engine-dir", str(tmp_path),
                          "--ip", "127.0.0.1", "--port", str(qs_port)],
                         os.environ)
        try:
            self._wait(qs, qs_port)
            req = urllib.request.Request(
                f"http://127.0.0.1:{qs_port}/queries.json",
                data=js.dumps({"x": 3}).encode(),
                headers={"Content-Type": "application/json"})
            assert js.loads(urllib.request.urlopen(req, timeout=5).read()
                            ) == {"result": 9}
        finally:
            qs.terminate(); qs.wait(timeout=10)
