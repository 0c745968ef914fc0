"""SDK client tests — run the real event + engine servers as subprocesses
on localhost ports (sharing the sqlite storage file via env) and drive
them with predictionio_amd.sdk exactly as an application using the
official PredictionIO Python SDK would."""

import os
import socket
import subprocess
import sys
import time

import pytest

from predictionio_amd.data.storage.base import AccessKey, App

HELPER = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "_sdk_server.py")


def _free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _wait_port(port: int, proc, timeout=20.0):
    t0 = time.time()
    while time.time() - t0 < timeout:
        if proc.poll() is not None:
            raise RuntimeError(
                f"server exited: {proc.stderr.read().decode()[:500]}")
        try:
            with socket.create_connection(("127.0.0.1", port), timeout=0.3):
                return
        except OSError:
            time.sleep(0.1)
    proc.terminate()
    raise RuntimeError("server did not come up")


def _spawn_server(kind: str, *extra) -> tuple:
    port = _free_port()
    root = os.path.dirname(os.path.dirname(HELPER))
    env = dict(os.environ)
    env["PYTHONPATH"] = root + os.pathsep + env.get("PYTHONPATH", "")
    proc = subprocess.Popen(
        [sys.executable, HELPER, kind, str(port), *extra],
        env=env, stdout=subprocess.DEVNULL, stderr=subprocess.PIPE,
        cwd=root)
    _wait_port(port, proc)
    return proc, f"http://127.0.0.1:{port}"


@pytest.fixture()
def event_server(mem_storage):
    app_id = mem_storage.get_meta_data_apps().insert(App(0, "sdkapp"))
    mem_storage.get_meta_data_access_keys().insert(
        AccessKey(key="SDKKEY", appid=app_id, events=[]))
    mem_storage.get_l_events().init(app_id)
    proc, url = _spawn_server("event")
    yield url
    proc.terminate()
    proc.wait(timeout=10)


class TestEventClient:
    def test_lifecycle(self, event_server):
        from predictionio_amd.sdk import (
            EventClient, NotCreatedError, NotFoundError,
        )
        bad = EventClient("WRONG", event_server)
        with pytest.raises(NotCreatedError):
            bad.set_user("u1")
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


class TestEngineClient:
    def test_query(self, mem_storage):
        from predictionio_amd.workflow import train as train_wf
        train_wf.run_train_from_variant({
            "id": "sdk", "engineFactory":
            "tests.fake_engine.JsonEngineFactory",
            "datasource": {"params": {"n": 4}},
            "algorithms": [{"name": "", "params": {}}]})
        proc, url = _spawn_server(
            "engine", "tests.fake_engine.JsonEngineFactory")
        try:
            from predictionio_amd.sdk import EngineClient
            ec = EngineClient(url)
            assert ec.send_query({"x": 2}) == {"result": 8}
        finally:
            proc.terminate()
            proc.wait(timeout=10)
