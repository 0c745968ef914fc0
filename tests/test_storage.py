"""Storage-layer contract tests (mirrors the reference's backend-agnostic
LEventsSpec / metadata specs shape — storage/jdbc/src/test/...)."""

from datetime import datetime, timedelta, timezone

import pytest

from predictionio_amd.data.events import DataMap, Event
from predictionio_amd.data.storage.base import (
    UNSET, AccessKey, App, Channel, EngineInstance, Model,
)


T0 = datetime(2026, 1, 1, tzinfo=timezone.utc)


def mk(event, eid, minutes=0, etype="user", **kw):
    return Event(event=event, entity_type=etype, entity_id=eid,
                 event_time=T0 + timedelta(minutes=minutes), **kw)


class TestMetadata:
    def test_apps_crud(self, mem_storage):
        apps = mem_storage.get_meta_data_apps()
        aid = apps.insert(App(0, "myapp", "desc"))
        assert aid
        assert apps.get(aid).name == "myapp"
        assert apps.get_by_name("myapp").id == aid
        assert apps.insert(App(0, "myapp")) is None  # duplicate name
        assert len(apps.get_all()) == 1
        a = apps.get(aid)
        a.description = "new"
        assert apps.update(a)
        assert apps.get(aid).description == "new"
        assert apps.delete(aid)
        assert apps.get(aid) is None

    def test_access_keys(self, mem_storage):
        ak = mem_storage.get_meta_data_access_keys()
        key = ak.insert(AccessKey("", 7, ["view", "buy"]))
        assert key
        got = ak.get(key)
        assert got.appid == 7 and got.events == ["view", "buy"]
        assert ak.get_by_app_id(7)[0].key == key
        assert ak.get_by_app_id(8) == []
        assert ak.delete(key)
        assert ak.get(key) is None

    def test_channels(self, mem_storage):
        ch = mem_storage.get_meta_data_channels()
        cid = ch.insert(Channel(0, "mobile", 7))
        assert cid
        assert ch.get(cid).name == "mobile"
        assert ch.insert(Channel(0, "bad name!", 7)) is None  # invalid name
        assert [c.name for c in ch.get_by_app_id(7)] == ["mobile"]
        assert ch.delete(cid)

    def test_engine_instances(self, mem_storage):
        ei = mem_storage.get_meta_data_engine_instances()
        rec = EngineInstance(
            id="", status="INIT", start_time=T0, end_time=T0,
            engine_id="e1", engine_version="1", engine_variant="default",
            engine_factory="pkg.Factory")
        iid = ei.insert(rec)
        assert ei.get(iid).status == "INIT"
        rec.id = iid
        rec.status = "COMPLETED"
        rec.start_time = T0 + timedelta(hours=1)
        assert ei.update(rec)
        # second completed, older
        rec2 = EngineInstance(
            id="", status="COMPLETED", start_time=T0, end_time=T0,
            engine_id="e1", engine_version="1", engine_variant="default",
            engine_factory="pkg.Factory")
        ei.insert(rec2)
        latest = ei.get_latest_completed("e1", "1", "default")
        assert latest.id == iid
        assert ei.get_latest_completed("e1", "2", "default") is None

    def test_models(self, mem_storage):
        m = mem_storage.get_model_data_models()
        m.insert(Model("i1", b"\x00\x01blob"))
        assert m.get("i1").models == b"\x00\x01blob"
        assert m.get("nope") is None
        assert m.delete("i1")


class TestLEvents:
    @pytest.fixture()
    def events(self, mem_storage):
        le = mem_storage.get_l_events()
        le.init(1)
        return le

    def test_insert_get_delete(self, events):
        eid = events.insert(mk("view", "u1"), 1)
        got = events.get(eid, 1)
        assert got.event == "view" and got.entity_id == "u1"
        assert events.delete(eid, 1)
        assert events.get(eid, 1) is None
        assert not events.delete(eid, 1)

    def test_insert_batch(self, events):
        ids = events.insert_batch([mk("view", f"u{i}") for i in range(5)], 1)
        assert len(set(ids)) == 5
        assert len(list(events.find(app_id=1))) == 5

    def test_find_filters(self, events):
        events.insert_batch([
            mk("view", "u1", 0, target_entity_type="item",
               target_entity_id="i1"),
            mk("buy", "u1", 1, target_entity_type="item",
               target_entity_id="i2"),
            mk("view", "u2", 2),
            mk("$set", "u1", 3, etype="item"),
        ], 1)
        assert len(list(events.find(app_id=1))) == 4
        assert len(list(events.find(app_id=1, entity_type="user"))) == 3
        assert len(list(events.find(app_id=1, entity_id="u1"))) == 3
        assert len(list(events.find(app_id=1, entity_type="user",
                                    entity_id="u1"))) == 2
        assert len(list(events.find(app_id=1, event_names=["view"]))) == 2
        assert len(list(events.find(app_id=1, event_names=["view", "buy"],
                                    entity_type="user"))) == 3
        # target entity: UNSET vs None vs value
        assert len(list(events.find(app_id=1, target_entity_type=None))) == 2
        assert len(list(events.find(app_id=1,
                                    target_entity_type="item"))) == 2
        assert len(list(events.find(app_id=1, target_entity_id="i2"))) == 1
        # time windows: [start, until)
        assert len(list(events.find(
            app_id=1, start_time=T0 + timedelta(minutes=1)))) == 3
        assert len(list(events.find(
            app_id=1, until_time=T0 + timedelta(minutes=1)))) == 1
        # limit + order
        lst = list(events.find(app_id=1, limit=2))
        assert [e.event_time for e in lst] == sorted(
            e.event_time for e in lst)
        rev = list(events.find(app_id=1, reversed=True, limit=1))
        assert rev[0].event_time == T0 + timedelta(minutes=3)

    def test_channels_separate(self, events):
        events.init(1, 5)
        events.insert(mk("view", "u1"), 1)
        events.insert(mk("buy", "u9"), 1, 5)
        assert [e.event for e in events.find(app_id=1)] == ["view"]
        assert [e.event for e in events.find(app_id=1, channel_id=5)] == ["buy"]
        events.remove(1, 5)
        assert list(events.find(app_id=1, channel_id=5)) == []

    def test_aggregate_properties(self, events):
        events.insert_batch([
            mk("$set", "i1", 0, etype="item",
               properties=DataMap({"color": "red", "price": 10})),
            mk("$set", "i1", 1, etype="item",
               properties=DataMap({"price": 12})),
            mk("$unset", "i1", 2, etype="item",
               properties=DataMap({"color": None})),
            mk("$set", "i2", 0, etype="item",
               properties=DataMap({"color": "blue"})),
            mk("view", "u1", 0),
        ], 1)
        agg = events.aggregate_properties(1, "item")
        assert agg["i1"].fields == {"price": 12}
        assert agg["i2"].fields == {"color": "blue"}
        req = events.aggregate_properties(1, "item", required=["color"])
        assert set(req) == {"i2"}


class TestEventStoreFacade:
    def test_app_name_resolution(self, mem_storage):
        from predictionio_amd.data import event_store
        apps = mem_storage.get_meta_data_apps()
        aid = apps.insert(App(0, "shop"))
        le = mem_storage.get_l_events()
        le.init(aid)
        le.insert(mk("view", "u1"), aid)
        evs = event_store.find(app_name="shop", entity_type="user")
        assert len(evs) == 1
        with pytest.raises(ValueError):
            event_store.find(app_name="nope")

    def test_find_by_entity(self, mem_storage):
        from predictionio_amd.data import event_store
        apps = mem_storage.get_meta_data_apps()
        aid = apps.insert(App(0, "shop2"))
        le = mem_storage.get_l_events()
        le.init(aid)
        le.insert_batch([mk("view", "u1", i) for i in range(3)], aid)
        evs = event_store.find_by_entity(
            app_name="shop2", entity_type="user", entity_id="u1",
            limit=2, latest=True, timeout=5.0)
        assert len(evs) == 2
        assert evs[0].event_time > evs[1].event_time


class TestModelsBackendMatrix:
    """Model-blob contract over every backend type (the reference runs its
    contract suites over backend combos via env — tests/run_docker.sh)."""

    @pytest.mark.parametrize("btype", ["sqlite", "localfs", "fsspec"])
    def test_blob_contract(self, btype, tmp_path, monkeypatch):
        from predictionio_amd.data import storage
        from predictionio_amd.data.storage.base import Model
        storage.reset()
        monkeypatch.setenv("PIO_FS_BASEDIR", str(tmp_path))
        monkeypatch.setenv("PIO_STORAGE_SOURCES_M_TYPE", btype)
        monkeypatch.setenv("PIO_STORAGE_SOURCES_M_PATH",
                           str(tmp_path / "blob"))
        monkeypatch.setenv("PIO_STORAGE_SOURCES_META_TYPE", "memory")
        for repo, src in (("METADATA", "META"), ("EVENTDATA", "META"),
                          ("MODELDATA", "M")):
            monkeypatch.setenv(f"PIO_STORAGE_REPOSITORIES_{repo}_SOURCE",
                               src)
            monkeypatch.setenv(f"PIO_STORAGE_REPOSITORIES_{repo}_NAME",
                               repo.lower())
        m = storage.get_model_data_models()
        blob = bytes(range(256)) * 10
        m.insert(Model("x1", blob))
        assert m.get("x1").models == blob
        m.insert(Model("x1", b"overwritten"))  # upsert semantics
        assert m.get("x1").models == b"overwritten"
        assert m.get("nope") is None
        assert m.delete("x1") is True
        assert m.delete("x1") is False
        storage.reset()


class TestConcurrentAccess:
    """The event server runs handlers on a thread pool (uvicorn): the
    sqlite DAO uses a connection per thread (file-backed) / one locked
    connection (:memory:) — hammer both from many threads."""

    def _hammer(self, le, app_id, threads=8, per=25):
        import threading
        from datetime import datetime, timezone
        from predictionio_amd.data.events import DataMap, Event
        errors = []

        def work(t):
            try:
                for k in range(per):
                    eid = le.insert(Event(
                        event="rate", entity_type="user",
                        entity_id=f"u{t}", target_entity_type="item",
                        target_entity_id=f"i{k}",
                        properties=DataMap({"rating": k % 5 + 1}),
                        event_time=datetime.now(timezone.utc)), app_id)
                    assert le.get(eid, app_id) is not None
                    le.find(app_id, entity_id=f"u{t}", limit=5)
            except Exception as e:  # noqa: BLE001
                errors.append(repr(e))

        ts = [threading.Thread(target=work, args=(t,))
              for t in range(threads)]
        for t in ts:
            t.start()
        for t in ts:
            t.join(timeout=60)
        assert not errors, errors[:3]
        evs = list(le.find(app_id, limit=-1))
        assert len(evs) == threads * per

    def test_sqlite_file_threaded(self, mem_storage):
        from predictionio_amd.data.storage.base import App
        app_id = mem_storage.get_meta_data_apps().insert(App(0, "ThrApp"))
        le = mem_storage.get_l_events()
        le.init(app_id)
        self._hammer(le, app_id)


class TestFindColumns:
    """Columnar bulk read (LEvents.find_columns): the sqlite store-side
    scan must agree with the base-class default (walking find())."""

    @pytest.fixture()
    def events(self, mem_storage):
        le = mem_storage.get_l_events()
        le.init(1)
        le.insert_batch([
            mk("rate", "u1", 0, target_entity_type="item",
               target_entity_id="i1",
               properties=DataMap({"rating": 4.5})),
            mk("buy", "u1", 1, target_entity_type="item",
               target_entity_id="i2"),
            mk("rate", "u2", 2, target_entity_type="item",
               target_entity_id="i1",
               properties=DataMap({"rating": 2.0})),
            mk("view", "u3", 3),
            mk("$set", "i1", 4, etype="item"),
        ], 1)
        return le

    def test_columns_match_find(self, events):
        from predictionio_amd.data.storage import base as sbase
        cols = events.find_columns(
            app_id=1, entity_type="user", event_names=["rate", "buy"],
            target_entity_type="item", property_fields=["rating"])
        ref = sbase.LEvents.find_columns(
            events, app_id=1, entity_type="user",
            event_names=["rate", "buy"], target_entity_type="item",
            property_fields=["rating"])
        assert cols["event"] == ref["event"] == ["rate", "buy", "rate"]
        assert cols["entity_id"] == ref["entity_id"] == ["u1", "u1", "u2"]
        assert (cols["target_entity_id"] == ref["target_entity_id"]
                == ["i1", "i2", "i1"])
        assert cols["rating"] == ref["rating"] == [4.5, None, 2.0]
        assert cols["event_time_ms"] == ref["event_time_ms"]

    def test_all_events_no_filters(self, events):
        cols = events.find_columns(app_id=1)
        assert len(cols["event"]) == 5
        # ascending event time
        assert cols["event_time_ms"] == sorted(cols["event_time_ms"])

    def test_bad_property_field_rejected(self, events):
        with pytest.raises(ValueError):
            events.find_columns(app_id=1,
                                property_fields=["x'); DROP TABLE t;--"])

    def test_channel_scoped_columns(self, mem_storage):
        le = mem_storage.get_l_events()
        le.init(2)
        le.init(2, 5)  # channel 5
        le.insert(mk("rate", "u1", 0, target_entity_type="item",
                     target_entity_id="i1",
                     properties=DataMap({"rating": 1.0})), 2)
        le.insert(mk("rate", "u2", 0, target_entity_type="item",
                     target_entity_id="i2",
                     properties=DataMap({"rating": 2.0})), 2, 5)
        default = le.find_columns(app_id=2, property_fields=["rating"])
        chan = le.find_columns(app_id=2, channel_id=5,
                               property_fields=["rating"])
        assert default["entity_id"] == ["u1"]
        assert chan["entity_id"] == ["u2"] and chan["rating"] == [2.0]
