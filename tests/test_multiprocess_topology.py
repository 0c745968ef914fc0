"""Full production process topology, each tier a REAL OS process:

    storage daemon  (pio storageserver)   ← owns the sqlite file
         ▲               ▲                      ▲
    event server     trainer (pio train)   query server (pio deploy)
    (pio eventserver)

all wired through the `remote` storage backend — the reference's
four-process operational shape (EventServer / CreateWorkflow driver /
CreateServer / storage tier, SURVEY §1 "Process topology") with the
client-server database the reference gets from PostgreSQL.
"""

import json
import os
import signal
import socket
import subprocess
import sys
import time
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
PIO = str(REPO / "bin" / "pio")


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _wait_http(url, timeout=30.0):
    import httpx
    t0 = time.time()
    while time.time() - t0 < timeout:
        try:
            r = httpx.get(url, timeout=2)
            if r.status_code < 500:
                return True
        except Exception:
            time.sleep(0.25)
    return False


@pytest.mark.timeout(300)
def test_four_process_lifecycle(tmp_path):
    import httpx

    sport = _free_port()
    eport = _free_port()
    qport = _free_port()

    # the DAEMON owns a sqlite file; every other process is a remote
    # client of it
    daemon_env = dict(
        os.environ,
        PIO_FS_BASEDIR=str(tmp_path),
        PIO_STORAGE_SOURCES_DB_TYPE="sqlite",
        PIO_STORAGE_SOURCES_DB_PATH=str(tmp_path / "served.sqlite"),
        PIO_STORAGE_REPOSITORIES_METADATA_SOURCE="DB",
        PIO_STORAGE_REPOSITORIES_METADATA_NAME="db",
        PIO_STORAGE_REPOSITORIES_EVENTDATA_SOURCE="DB",
        PIO_STORAGE_REPOSITORIES_EVENTDATA_NAME="db",
        PIO_STORAGE_REPOSITORIES_MODELDATA_SOURCE="DB",
        PIO_STORAGE_REPOSITORIES_MODELDATA_NAME="db",
    )
    client_env = dict(
        os.environ,
        PIO_FS_BASEDIR=str(tmp_path),
        PIO_STORAGE_SOURCES_RS_TYPE="remote",
        PIO_STORAGE_SOURCES_RS_URL=f"http://127.0.0.1:{sport}",
        PIO_STORAGE_REPOSITORIES_METADATA_SOURCE="RS",
        PIO_STORAGE_REPOSITORIES_METADATA_NAME="rs",
        PIO_STORAGE_REPOSITORIES_EVENTDATA_SOURCE="RS",
        PIO_STORAGE_REPOSITORIES_EVENTDATA_NAME="rs",
        PIO_STORAGE_REPOSITORIES_MODELDATA_SOURCE="RS",
        PIO_STORAGE_REPOSITORIES_MODELDATA_NAME="rs",
    )

    procs = []

    def spawn(args, env):
        p = subprocess.Popen(args, env=env, stdout=subprocess.PIPE,
                             stderr=subprocess.STDOUT, text=True,
                             cwd=str(REPO))
        procs.append(p)
        return p

    try:
        spawn([PIO, "storageserver", "--ip", "127.0.0.1",
               "--port", str(sport)], daemon_env)
        assert _wait_http(f"http://127.0.0.1:{sport}/"), "daemon up"

        # app + key through the remote backend (CLI is a client process)
        r = subprocess.run(
            [PIO, "app", "new", "TopoApp", "--access-key", "topokey"],
            env=client_env, capture_output=True, text=True, cwd=str(REPO))
        assert r.returncode == 0, r.stdout + r.stderr

        spawn([PIO, "eventserver", "--ip", "127.0.0.1",
               "--port", str(eport)], client_env)
        assert _wait_http(f"http://127.0.0.1:{eport}/"), "eventserver up"

        # ingest rate events over HTTP (client #2)
        with httpx.Client(base_url=f"http://127.0.0.1:{eport}",
                          timeout=30) as c:
            batch = []
            for u in range(30):
                for j in range(6):
                    item = (u + 2 * j) % 15
                    batch.append({
                        "event": "rate", "entityType": "user",
                        "entityId": f"u{u}",
                        "targetEntityType": "item",
                        "targetEntityId": f"i{item}",
                        "properties": {
                            "rating": 5.0 if (u + item) % 2 == 0
                            else 1.0}})
            for i in range(0, len(batch), 50):
                r = c.post("/batch/events.json?accessKey=topokey",
                           json=batch[i:i + 50])
                assert r.status_code == 200
                assert all(x["status"] == 201 for x in r.json())

        # train as its own process (pio train — the engine template dir)
        engine_dir = tmp_path / "engine"
        engine_dir.mkdir()
        (engine_dir / "engine.json").write_text(json.dumps({
            "id": "topo", "version": "1",
            "engineFactory":
                "predictionio_amd.templates.recommendation"
                ".RecommendationEngine",
            "datasource": {"params": {"appName": "TopoApp"}},
            "algorithms": [{"name": "als", "params": {
                "rank": 8, "numIterations": 5, "lambda": 0.1,
                "seed": 1}}],
        }))
        r = subprocess.run([PIO, "train"], env=client_env,
                           capture_output=True, text=True,
                           cwd=str(engine_dir), timeout=180)
        assert r.returncode == 0, r.stdout + r.stderr

        # deploy as its own process, then query over HTTP
        spawn([PIO, "deploy", "--ip", "127.0.0.1", "--port", str(qport),
               "--engine-dir", str(engine_dir)], client_env)
        assert _wait_http(f"http://127.0.0.1:{qport}/", timeout=60), \
            "query server up"
        with httpx.Client(base_url=f"http://127.0.0.1:{qport}",
                          timeout=30) as c:
            r = c.post("/queries.json", json={"user": "u2", "num": 4})
            assert r.status_code == 200, r.text
            scores = r.json()["itemScores"]
            assert len(scores) == 4
            assert all("item" in s and "score" in s for s in scores)
    finally:
        for p in procs:
            p.send_signal(signal.SIGTERM)
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
