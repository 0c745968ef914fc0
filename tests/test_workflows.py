"""Workflow-layer tests: evaluation instances, FastEvalEngine caching,
batch predict, import/export, self-cleaning, dashboard + admin servers.

Mirrors the roles of FastEvalEngineTest.scala (stage-cache hit counts),
EvaluationWorkflow, BatchPredict, FileToEvents/EventsToFile, and
SelfCleaningDataSource behaviors."""

import json

import pytest
from fastapi.testclient import TestClient

from predictionio_amd.controller import EngineParams, Params
from tests.fake_engine import DS0, Prep0, Algo0, Serve0, make_params


class TestFastEvalEngine:
    def _engine(self):
        from predictionio_amd.controller.fast_eval import FastEvalEngine
        return FastEvalEngine(DS0, Prep0, {"": Algo0}, Serve0)

    def test_cache_hits(self):
        e = self._engine()
        # 3 candidates sharing ds+prep, differing algo params
        eps = [EngineParams(data_source_params=Params({"n": 4}),
                            preparator_params=Params({"scale": 2}),
                            algorithms_params=[("", Params({"bias": b}))])
               for b in (0, 1, 2)]
        results = e.batch_eval(eps)
        assert len(results) == 3
        # datasource + preparator computed once; the prep-level cache
        # short-circuits, so reuse shows up as preparator hits
        assert e.misses["datasource"] == 1
        assert e.misses["preparator"] == 1 and e.hits["preparator"] == 2
        assert e.misses["algorithms"] == 3
        # distinct ds params → new datasource read
        e.batch_eval([EngineParams(
            data_source_params=Params({"n": 6}),
            preparator_params=Params({"scale": 2}),
            algorithms_params=[("", Params())])])
        assert e.misses["datasource"] == 2

    def test_matches_plain_engine(self):
        from predictionio_amd.controller import Engine
        plain = Engine(DS0, Prep0, {"": Algo0}, Serve0)
        fast = self._engine()
        ep = EngineParams(data_source_params=Params({"n": 4}),
                          algorithms_params=[("", Params())])
        assert plain.eval(ep) == fast.eval(ep)


class TestEvaluationWorkflow:
    def test_records_instance(self, mem_storage):
        from predictionio_amd.controller import Evaluation, ZeroMetric
        from predictionio_amd.controller import Engine
        from predictionio_amd.workflow.evaluation import run_evaluation
        e = Engine(DS0, Prep0, {"": Algo0}, Serve0)
        ev = Evaluation(engine=e, metric=ZeroMetric())
        ep = EngineParams(algorithms_params=[("", Params())])
        iid, result = run_evaluation(ev, [ep], evaluation_class="Zero")
        inst = mem_storage.get_meta_data_evaluation_instances().get(iid)
        assert inst.status == "EVALCOMPLETED"
        assert "Zero" in inst.evaluation_class
        assert json.loads(inst.evaluator_results_json)["bestScore"] == 0.0

    def test_failed_marks_instance(self, mem_storage):
        from predictionio_amd.controller import Evaluation, ZeroMetric
        from predictionio_amd.controller import Engine
        from predictionio_amd.workflow.evaluation import run_evaluation

        class BoomDS(DS0):
            def read_eval(self):
                raise RuntimeError("boom")

        e = Engine(BoomDS, Prep0, {"": Algo0}, Serve0)
        ev = Evaluation(engine=e, metric=ZeroMetric())
        with pytest.raises(RuntimeError):
            run_evaluation(ev, [EngineParams(
                algorithms_params=[("", Params())])])
        insts = mem_storage.get_meta_data_evaluation_instances().get_all()
        assert insts and insts[-1].status == "FAILED"


class TestBatchPredict:
    def test_file_roundtrip(self, mem_storage, tmp_path):
        from predictionio_amd.workflow import train as train_wf
        from predictionio_amd.workflow.batch_predict import run_batch_predict
        variant = {"id": "bp", "engineFactory":
                   "tests.fake_engine.JsonEngineFactory",
                   "datasource": {"params": {"n": 4}},
                   "algorithms": [{"name": "", "params": {}}]}
        train_wf.run_train_from_variant(variant)
        qfile = tmp_path / "queries.json"
        qfile.write_text('{"x": 1}\n{"x": 2}\n\n{"x": 3}\n')
        out = tmp_path / "out.json"
        n = run_batch_predict(variant, str(qfile), str(out))
        assert n == 3
        lines = [json.loads(l) for l in out.read_text().splitlines()]
        assert lines[0]["prediction"] == {"result": 7}  # sum(0..3)+1
        assert lines[2]["query"] == {"x": 3}


class TestImportExport:
    def test_roundtrip(self, mem_storage, tmp_path):
        from predictionio_amd.data.storage.base import App
        from predictionio_amd.workflow.import_export import (
            export_events, import_events,
        )
        app_id = mem_storage.get_meta_data_apps().insert(App(0, "imp"))
        f = tmp_path / "events.json"
        evs = [{"event": "rate", "entityType": "user", "entityId": f"u{i}",
                "targetEntityType": "item", "targetEntityId": "i1",
                "properties": {"rating": i},
                "eventTime": f"2026-01-0{i+1}T00:00:00.000Z"}
               for i in range(3)]
        f.write_text("\n".join(json.dumps(e) for e in evs))
        assert import_events(app_id, str(f)) == 3
        out = tmp_path / "export.json"
        assert export_events(app_id, str(out)) == 3
        dumped = [json.loads(l) for l in out.read_text().splitlines()]
        assert {d["entityId"] for d in dumped} == {"u0", "u1", "u2"}

    def test_import_validates(self, mem_storage, tmp_path):
        from predictionio_amd.data.storage.base import App
        from predictionio_amd.workflow.import_export import import_events
        app_id = mem_storage.get_meta_data_apps().insert(App(0, "imp2"))
        f = tmp_path / "bad.json"
        f.write_text(json.dumps({"event": "", "entityType": "u",
                                 "entityId": "x"}))
        with pytest.raises(Exception):
            import_events(app_id, str(f))


class TestSelfCleaning:
    def _seed(self, storage, app_id):
        from datetime import timedelta
        from predictionio_amd.data.events import DataMap, Event, utcnow
        le = storage.get_l_events()
        old = utcnow() - timedelta(seconds=7200)
        le.insert(Event(event="view", entity_type="user", entity_id="u1",
                        target_entity_type="item", target_entity_id="i1",
                        event_time=old), app_id)
        for _ in range(2):  # duplicates
            le.insert(Event(event="view", entity_type="user",
                            entity_id="u2", target_entity_type="item",
                            target_entity_id="i2",
                            event_time=utcnow()), app_id)
        le.insert(Event(event="$set", entity_type="user", entity_id="u3",
                        properties=DataMap({"a": 1}),
                        event_time=utcnow() - timedelta(seconds=10)),
                  app_id)
        le.insert(Event(event="$set", entity_type="user", entity_id="u3",
                        properties=DataMap({"b": 2}),
                        event_time=utcnow()), app_id)

    def test_clean(self, mem_storage):
        from predictionio_amd.controller.self_cleaning import (
            EventWindow, SelfCleaningDataSource,
        )
        from predictionio_amd.data.storage.base import App
        app_id = mem_storage.get_meta_data_apps().insert(App(0, "clean"))
        mem_storage.get_l_events().init(app_id)
        self._seed(mem_storage, app_id)

        class DS(SelfCleaningDataSource):
            app_name = "clean"
            event_window = EventWindow(duration=3600.0,
                                       remove_duplicates=True,
                                       compress_properties=True)

        ds = DS()
        cleaned = ds.read_cleaned_events()
        # old event dropped, dup removed, two $set folded into one
        assert len(cleaned) == 2
        sets = [e for e in cleaned if e.event == "$set"]
        assert len(sets) == 1
        assert sets[0].properties.get("a") == 1
        assert sets[0].properties.get("b") == 2
        # persisted rewrite
        kept = ds.clean_persisted_events()
        assert kept == 2
        left = list(mem_storage.get_l_events().find(app_id))
        assert len(left) == 2


class TestDashboardAdmin:
    def test_dashboard_lists_eval(self, mem_storage):
        from predictionio_amd.controller import (
            Engine, Evaluation, ZeroMetric,
        )
        from predictionio_amd.workflow.evaluation import run_evaluation
        e = Engine(DS0, Prep0, {"": Algo0}, Serve0)
        iid, _ = run_evaluation(Evaluation(engine=e, metric=ZeroMetric()),
                                [EngineParams(
                                    algorithms_params=[("", Params())])],
                                evaluation_class="ZeroEval")
        from predictionio_amd.server.dashboard import create_app
        c = TestClient(create_app())
        assert iid in c.get("/").text
        assert c.get(f"/engine_instances/{iid}").status_code == 200
        body = c.get(f"/engine_instances/{iid}/evaluator_results.json")
        assert body.json()["bestScore"] == 0.0

    def test_admin_app_lifecycle(self, mem_storage):
        from predictionio_amd.server.admin import create_app
        c = TestClient(create_app())
        assert c.get("/").json() == {"status": "alive"}
        r = c.post("/cmd/app", json={"name": "adminapp"})
        assert r.status_code == 200 and "accessKey" in r.json()
        assert c.post("/cmd/app",
                      json={"name": "adminapp"}).status_code == 409
        apps = c.get("/cmd/app").json()["apps"]
        assert any(a["name"] == "adminapp" for a in apps)
        assert c.delete("/cmd/app/adminapp/data").status_code == 200
        assert c.delete("/cmd/app/adminapp").status_code == 200
        assert c.delete("/cmd/app/adminapp").status_code == 404


class TestAdminExtended:
    def test_keys_channels_status(self, mem_storage):
        from fastapi.testclient import TestClient

        from predictionio_amd.server.admin import create_app
        c = TestClient(create_app())
        assert c.get("/status").json()["status"] == "ok"
        r = c.post("/cmd/app", json={"name": "extapp"})
        assert r.status_code == 200
        # access keys
        k = c.post("/cmd/app/extapp/accesskey",
                   json={"events": ["view"]}).json()["accessKey"]
        detail = c.get("/cmd/app/extapp").json()
        assert any(e["key"] == k and e["events"] == ["view"]
                   for e in detail["accessKeys"])
        assert c.delete(f"/cmd/accesskey/{k}").status_code == 200
        assert c.delete(f"/cmd/accesskey/{k}").status_code == 404
        # channels
        r = c.post("/cmd/app/extapp/channel", json={"name": "live"})
        assert r.status_code == 200
        assert c.post("/cmd/app/extapp/channel",
                      json={"name": "bad name!"}).status_code == 400
        assert any(ch["name"] == "live"
                   for ch in c.get("/cmd/app/extapp").json()["channels"])
        assert c.delete("/cmd/app/extapp/channel/live").status_code == 200
        assert c.delete("/cmd/app/extapp/channel/live").status_code == 404
        # engine instances listing
        assert c.get("/cmd/engineinstances").json()["engineInstances"] == []


class TestDashboardTraining:
    def test_training_runs_listed(self, mem_storage):
        from fastapi.testclient import TestClient

        from predictionio_amd.data.storage.base import (
            EngineInstance, utcnow,
        )
        from predictionio_amd.server.dashboard import create_app
        ei = mem_storage.get_meta_data_engine_instances()
        iid = ei.insert(EngineInstance(
            id="", status="COMPLETED", start_time=utcnow(),
            end_time=utcnow(), engine_id="e", engine_version="1",
            engine_variant="engine.json", engine_factory="MyFactory",
            algorithms_params="[('als', {...})]"))
        c = TestClient(create_app())
        html = c.get("/").text
        assert "MyFactory" in html and iid in html
        detail = c.get(f"/training/{iid}").text
        assert "MyFactory" in detail and "algorithmsParams" in detail
        assert c.get("/training/nope").status_code == 404
