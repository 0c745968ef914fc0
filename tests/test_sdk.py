"""SDK client tests — run the real event + engine servers on localhost
ports (uvicorn in background threads) and drive them with
predictionio_amd.sdk exactly as an application using the official
PredictionIO Python SDK would."""

import socket
import threading
import time

import pytest

from predictionio_amd.data.storage.base import AccessKey, App


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _serve(app, port):
    import uvicorn
    cfg = uvicorn.Config(app, host="127.0.0.1", port=port,
                         log_level="error")
    server = uvicorn.Server(cfg)
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    for _ in range(100):
        if server.started:
            return server
        time.sleep(0.05)
    raise RuntimeError("server did not start")


@pytest.fixture()
def event_server(mem_storage):
    app_id = mem_storage.get_meta_data_apps().insert(App(0, "sdkapp"))
    mem_storage.get_meta_data_access_keys().insert(
        AccessKey(key="SDKKEY", appid=app_id, events=[]))
    mem_storage.get_l_events().init(app_id)
    from predictionio_amd.server.eventserver import create_app
    port = _free_port()
    server = _serve(create_app(), port)
    yield f"http://127.0.0.1:{port}"
    server.should_exit = True
    time.sleep(0.1)


class TestEventClient:
    def test_lifecycle(self, event_server):
        from predictionio_amd.sdk import EventClient, NotFoundError
        c = EventClient("SDKKEY", event_server)
        r = c.record_user_action_on_item("rate", "u1", "i1",
                                         {"rating": 4.0})
        eid = r["eventId"]
        got = c.get_event(eid)
        assert got["event"] == "rate" and got["targetEntityId"] == "i1"
        c.set_user("u1", {"age": 30})
        c.set_item("i1", {"categories": ["a"]})
        evs = c.get_events(event="$set", limit=-1)
        assert len(evs) == 2
        assert c.delete_event(eid) == {"message": "Found"}
        with pytest.raises(NotFoundError):
            c.get_event(eid)

    def test_bad_key_raises(self, event_server):
        from predictionio_amd.sdk import EventClient, NotCreatedError
        c = EventClient("WRONG", event_server)
        with pytest.raises(NotCreatedError):
            c.set_user("u1")


class TestEngineClient:
    def test_query(self, mem_storage):
        from predictionio_amd.workflow import train as train_wf
        train_wf.run_train_from_variant({
            "id": "sdk", "engineFactory":
            "tests.fake_engine.JsonEngineFactory",
            "datasource": {"params": {"n": 4}},
            "algorithms": [{"name": "", "params": {}}]})
        from predictionio_amd.server.queryserver import (
            ServerConfig, create_app,
        )
        port = _free_port()
        server = _serve(create_app(ServerConfig(
            engine_factory="tests.fake_engine.JsonEngineFactory")), port)
        try:
            from predictionio_amd.sdk import EngineClient
            ec = EngineClient(f"http://127.0.0.1:{port}")
            assert ec.send_query({"x": 2}) == {"result": 8}
        finally:
            server.should_exit = True
            time.sleep(0.1)
