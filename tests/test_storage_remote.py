"""Contract tests for the `remote` (client-server) storage backend.

A storage-server daemon (uvicorn thread, sqlite-file backed via
create_app(sqlite_path=...)) serves the DAO surface; the test process is
configured with PIO_STORAGE_SOURCES_*_TYPE=remote, so every call below
goes client → HTTP → daemon → sqlite — the reference's backend-matrix
shape (tests/run_docker.sh runs the same suites over PGSQL/MYSQL/...).
"""

import socket
import threading
import time
from datetime import datetime, timedelta, timezone

import pytest

from predictionio_amd.data.events import DataMap, Event
from predictionio_amd.data.storage.base import (
    AccessKey, App, Channel, EngineInstance, Model, utcnow,
)

T0 = datetime(2021, 6, 1, tzinfo=timezone.utc)


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


@pytest.fixture(scope="module")
def storage_server(tmp_path_factory):
    import uvicorn

    from predictionio_amd.server.storageserver import create_app
    tmp = tmp_path_factory.mktemp("remote_store")
    port = _free_port()
    app = create_app(sqlite_path=str(tmp / "served.sqlite"))
    cfg = uvicorn.Config(app, host="127.0.0.1", port=port,
                         log_level="error")
    server = uvicorn.Server(cfg)
    th = threading.Thread(target=server.run, daemon=True)
    th.start()
    import httpx
    for _ in range(100):
        try:
            httpx.get(f"http://127.0.0.1:{port}/", timeout=1)
            break
        except Exception:
            time.sleep(0.1)
    yield port
    server.should_exit = True
    th.join(timeout=5)


@pytest.fixture()
def remote_storage(storage_server, monkeypatch):
    from predictionio_amd.data import storage
    storage.reset()
    monkeypatch.setenv("PIO_STORAGE_SOURCES_RS_TYPE", "remote")
    monkeypatch.setenv("PIO_STORAGE_SOURCES_RS_URL",
                       f"http://127.0.0.1:{storage_server}")
    for repo in ("METADATA", "EVENTDATA", "MODELDATA"):
        monkeypatch.setenv(f"PIO_STORAGE_REPOSITORIES_{repo}_SOURCE", "RS")
        monkeypatch.setenv(f"PIO_STORAGE_REPOSITORIES_{repo}_NAME", "rs")
    yield storage
    storage.reset()


def mk(event, eid, minutes=0, etype="user", **kw):
    return Event(event=event, entity_type=etype, entity_id=eid,
                 event_time=T0 + timedelta(minutes=minutes), **kw)


class TestRemoteMetadata:
    def test_apps(self, remote_storage):
        apps = remote_storage.get_meta_data_apps()
        aid = apps.insert(App(0, "remoteapp", "d"))
        assert aid and apps.get(aid).name == "remoteapp"
        assert apps.get_by_name("remoteapp").id == aid
        assert apps.insert(App(0, "remoteapp")) is None  # dup name
        assert any(a.id == aid for a in apps.get_all())
        assert apps.update(App(aid, "remoteapp", "d2"))
        assert apps.get(aid).description == "d2"
        assert apps.delete(aid)
        assert apps.get(aid) is None

    def test_access_keys_and_channels(self, remote_storage):
        apps = remote_storage.get_meta_data_apps()
        aid = apps.insert(App(0, "rkapp"))
        ak = remote_storage.get_meta_data_access_keys()
        key = ak.insert(AccessKey("", aid, ["view"]))
        assert key
        got = ak.get(key)
        assert got.appid == aid and got.events == ["view"]
        assert [k.key for k in ak.get_by_app_id(aid)] == [key]
        ch = remote_storage.get_meta_data_channels()
        cid = ch.insert(Channel(0, "live", aid))
        assert cid and ch.get(cid).name == "live"
        assert [c.id for c in ch.get_by_app_id(aid)] == [cid]
        assert ch.delete(cid)
        assert ak.delete(key)

    def test_engine_instances(self, remote_storage):
        ei = remote_storage.get_meta_data_engine_instances()
        now = utcnow()
        iid = ei.insert(EngineInstance(
            id="", status="INIT", start_time=now, end_time=now,
            engine_id="e1", engine_version="1", engine_variant="v",
            engine_factory="F1"))
        inst = ei.get(iid)
        assert inst.status == "INIT" and inst.engine_factory == "F1"
        inst.status = "COMPLETED"
        assert ei.update(inst)
        assert ei.get_latest_completed_by_factory("F1").id == iid
        assert ei.get_latest_completed("e1", "1", "v").id == iid
        assert ei.get_completed("e1", "1", "v")[0].id == iid

    def test_models_blob(self, remote_storage):
        models = remote_storage.get_model_data_models()
        blob = bytes(range(256)) * 10
        models.insert(Model(id="m1", models=blob))
        assert models.get("m1").models == blob
        assert models.delete("m1")
        assert models.get("m1") is None


class TestRemoteEvents:
    @pytest.fixture()
    def events(self, remote_storage):
        le = remote_storage.get_l_events()
        le.init(7)
        yield le
        le.remove(7)

    def test_insert_get_find_delete(self, events):
        eid = events.insert(
            mk("rate", "u1", 0, target_entity_type="item",
               target_entity_id="i1",
               properties=DataMap({"rating": 4.0})), 7)
        got = events.get(eid, 7)
        assert got.event == "rate" and got.properties.get("rating") == 4.0
        ids = events.insert_batch(
            [mk("buy", "u1", 1, target_entity_type="item",
                target_entity_id="i2"),
             mk("view", "u2", 2)], 7)
        assert len(ids) == 2
        assert len(list(events.find(app_id=7))) == 3
        assert len(list(events.find(app_id=7, entity_id="u1"))) == 2
        assert len(list(events.find(app_id=7, event_names=["buy"]))) == 1
        assert len(list(events.find(app_id=7,
                                    target_entity_type=None))) == 1
        # reversed ordering
        evs = list(events.find(app_id=7, reversed=True))
        assert evs[0].event == "view"
        assert events.delete(eid, 7)
        assert events.get(eid, 7) is None

    def test_find_columns(self, events):
        events.insert_batch([
            mk("rate", "u1", 0, target_entity_type="item",
               target_entity_id="i1",
               properties=DataMap({"rating": 4.5})),
            mk("rate", "u2", 1, target_entity_type="item",
               target_entity_id="i2",
               properties=DataMap({"rating": 2.0})),
        ], 7)
        cols = events.find_columns(app_id=7, entity_type="user",
                                   event_names=["rate"],
                                   property_fields=["rating"])
        assert cols["entity_id"] == ["u1", "u2"]
        assert cols["target_entity_id"] == ["i1", "i2"]
        assert cols["rating"] == [4.5, 2.0]

    def test_aggregate_properties(self, events):
        events.insert_batch([
            mk("$set", "i1", 0, etype="item",
               properties=DataMap({"price": 10, "cat": "a"})),
            mk("$set", "i1", 1, etype="item",
               properties=DataMap({"price": 12})),
            mk("$unset", "i1", 2, etype="item",
               properties=DataMap({"cat": None})),
        ], 7)
        props = events.aggregate_properties(app_id=7, entity_type="item")
        assert props["i1"].get("price") == 12
        assert "cat" not in props["i1"].fields


class TestRemoteEndToEnd:
    def test_template_trains_through_remote_store(self, remote_storage):
        """The recommendation template's full columnar train path over
        the client-server backend."""
        apps = remote_storage.get_meta_data_apps()
        aid = apps.insert(App(0, "RemoteTrainApp"))
        le = remote_storage.get_l_events()
        le.init(aid)
        evs = []
        for u in range(30):
            for i in range(8):
                item = (u + 2 * i) % 20
                evs.append(Event(
                    event="rate", entity_type="user", entity_id=f"u{u}",
                    target_entity_type="item", target_entity_id=f"i{item}",
                    properties=DataMap(
                        {"rating": 5.0 if (u + item) % 2 == 0 else 1.0}),
                    event_time=T0 + timedelta(minutes=u * 10 + i)))
        le.insert_batch(evs, aid)
        from predictionio_amd.controller import EngineParams, Params
        from predictionio_amd.templates.recommendation import (
            ALSAlgorithm, Query, RecommendationEngine,
        )
        e = RecommendationEngine.apply()
        ep = EngineParams(
            data_source_params=Params({"appName": "RemoteTrainApp"}),
            algorithms_params=[("als", Params(
                {"rank": 8, "numIterations": 5, "lambda": 0.1,
                 "seed": 1}))])
        models = e.train(ep)
        algo = ALSAlgorithm(ep.algorithms_params[0][1])
        r = algo.predict(models[0], Query(user="u2", num=5))
        assert len(r.item_scores) == 5


class TestRemoteConcurrency:
    def test_concurrent_multi_client_ingest_and_reads(self, remote_storage):
        """16 client threads hammer one storage daemon with mixed
        writes/reads — the concurrent multi-process access class the
        reference gets from PostgreSQL."""
        import concurrent.futures

        from predictionio_amd.data.storage.base import App
        apps = remote_storage.get_meta_data_apps()
        aid = apps.insert(App(0, "ConcApp"))
        le = remote_storage.get_l_events()
        le.init(aid)

        def worker(w):
            n_ok = 0
            for i in range(10):
                evs = [mk("view", f"w{w}u{i}_{j}", minutes=w * 100 + i,
                          target_entity_type="item",
                          target_entity_id=f"i{j}") for j in range(10)]
                ids = le.insert_batch(evs, aid)
                n_ok += len(ids)
                # interleaved reads
                got = list(le.find(app_id=aid, entity_id=f"w{w}u{i}_0"))
                assert len(got) == 1
            return n_ok

        with concurrent.futures.ThreadPoolExecutor(16) as pool:
            totals = list(pool.map(worker, range(16)))
        assert sum(totals) == 16 * 10 * 10
        cols = le.find_columns(app_id=aid, event_names=["view"])
        assert len(cols["event"]) == 1600
        le.remove(aid)
