"""Storage server daemon — the server side of the `remote` backend.

One long-running process owns the database (whatever the DAEMON's own
PIO_STORAGE_* env selects — sqlite WAL file by default) and serves the
full DAO contract over HTTP to every other framework process (Event
Server, trainer, query server, dashboard). This is the framework's
equivalent of the reference's production client-server database tier
(PostgreSQL behind JDBC, storage/jdbc/.../JDBCLEvents.scala:55-88):
concurrent multi-process access goes through ONE server, not N
processes opening one SQLite file.

Run: `pio storageserver --port 7072` (or python -m uvicorn ...); point
clients at it with
    PIO_STORAGE_SOURCES_REMOTE_TYPE=remote
    PIO_STORAGE_SOURCES_REMOTE_URL=http://host:7072
"""

from __future__ import annotations

import base64
import logging
from typing import Any, Dict

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse

from predictionio_amd.data.events import Event
from predictionio_amd.data.storage.base import (
    AccessKey, App, Channel, Model,
)
from predictionio_amd.data.storage.remote import (
    ei_from_dict, ei_to_dict, evi_from_dict, evi_to_dict, from_ms, ms,
    opt_unfield,
)

logger = logging.getLogger(__name__)


def create_app(sqlite_path: str = None) -> FastAPI:
    """sqlite_path: serve a specific sqlite file directly (bypasses the
    env registry — used when the daemon's own env must not be the
    `remote` type it is serving, e.g. under test, or to pin the file).
    Default: the daemon's own PIO_STORAGE_* configuration."""
    app = FastAPI(title="predictionio_amd storage server")

    if sqlite_path is not None:
        from predictionio_amd.data.storage import sqlite as sq
        client = sq.SQLiteClient(sqlite_path)
        daos = {
            "apps": sq.SQLiteApps(client),
            "accesskeys": sq.SQLiteAccessKeys(client),
            "channels": sq.SQLiteChannels(client),
            "engineinstances": sq.SQLiteEngineInstances(client),
            "evaluationinstances": sq.SQLiteEvaluationInstances(client),
            "models": sq.SQLiteModels(client),
            "levents": sq.SQLiteLEvents(client),
        }

        def dao_of(kind):
            return daos[kind]
    else:
        from predictionio_amd.data import storage

        def dao_of(kind):
            return {
                "apps": storage.get_meta_data_apps,
                "accesskeys": storage.get_meta_data_access_keys,
                "channels": storage.get_meta_data_channels,
                "engineinstances": storage.get_meta_data_engine_instances,
                "evaluationinstances":
                    storage.get_meta_data_evaluation_instances,
                "models": storage.get_model_data_models,
                "levents": storage.get_l_events,
            }[kind]()

    def handle(kind: str, method: str, p: Dict[str, Any]) -> Any:
        if kind == "apps":
            dao = dao_of("apps")
            if method == "insert":
                return dao.insert(App(p.get("id") or 0, p["name"],
                                      p.get("description")))
            if method == "get":
                a = dao.get(p["id"])
                return a.__dict__ if a else None
            if method == "get_by_name":
                a = dao.get_by_name(p["name"])
                return a.__dict__ if a else None
            if method == "get_all":
                return [a.__dict__ for a in dao.get_all()]
            if method == "update":
                return dao.update(App(p["id"], p["name"],
                                      p.get("description")))
            if method == "delete":
                return dao.delete(p["id"])
        elif kind == "accesskeys":
            dao = dao_of("accesskeys")
            if method == "insert":
                return dao.insert(AccessKey(p["key"], p["appid"],
                                            p.get("events") or []))
            if method == "get":
                k = dao.get(p["key"])
                return k.__dict__ if k else None
            if method == "get_all":
                return [k.__dict__ for k in dao.get_all()]
            if method == "get_by_app_id":
                return [k.__dict__ for k in dao.get_by_app_id(p["appid"])]
            if method == "update":
                return dao.update(AccessKey(p["key"], p["appid"],
                                            p.get("events") or []))
            if method == "delete":
                return dao.delete(p["key"])
        elif kind == "channels":
            dao = dao_of("channels")
            if method == "insert":
                return dao.insert(Channel(p.get("id") or 0, p["name"],
                                          p["appid"]))
            if method == "get":
                c = dao.get(p["id"])
                return c.__dict__ if c else None
            if method == "get_by_app_id":
                return [c.__dict__ for c in dao.get_by_app_id(p["appid"])]
            if method == "delete":
                return dao.delete(p["id"])
        elif kind == "engineinstances":
            dao = dao_of("engineinstances")
            if method == "insert":
                return dao.insert(ei_from_dict(p))
            if method == "get":
                i = dao.get(p["id"])
                return ei_to_dict(i) if i else None
            if method == "get_all":
                return [ei_to_dict(i) for i in dao.get_all()]
            if method == "get_completed":
                return [ei_to_dict(i) for i in dao.get_completed(
                    p["engineId"], p["engineVersion"], p["engineVariant"])]
            if method == "get_latest_completed":
                i = dao.get_latest_completed(
                    p["engineId"], p["engineVersion"], p["engineVariant"])
                return ei_to_dict(i) if i else None
            if method == "get_latest_completed_by_factory":
                i = dao.get_latest_completed_by_factory(
                    p["engineFactory"], p.get("engineVariant"))
                return ei_to_dict(i) if i else None
            if method == "update":
                return dao.update(ei_from_dict(p))
            if method == "delete":
                return dao.delete(p["id"])
        elif kind == "evaluationinstances":
            dao = dao_of("evaluationinstances")
            if method == "insert":
                return dao.insert(evi_from_dict(p))
            if method == "get":
                i = dao.get(p["id"])
                return evi_to_dict(i) if i else None
            if method == "get_all":
                return [evi_to_dict(i) for i in dao.get_all()]
            if method == "get_completed":
                return [evi_to_dict(i) for i in dao.get_completed()]
            if method == "update":
                return dao.update(evi_from_dict(p))
            if method == "delete":
                return dao.delete(p["id"])
        elif kind == "models":
            dao = dao_of("models")
            if method == "insert":
                dao.insert(Model(id=p["id"],
                                 models=base64.b64decode(p["models"])))
                return True
            if method == "get":
                m = dao.get(p["id"])
                if m is None:
                    return None
                return {"id": m.id,
                        "models": base64.b64encode(m.models).decode()}
            if method == "delete":
                return dao.delete(p["id"])
        elif kind == "levents":
            dao = dao_of("levents")
            app_id = p.get("appId")
            ch = p.get("channelId")
            if method == "init":
                return dao.init(app_id, ch)
            if method == "remove":
                return dao.remove(app_id, ch)
            if method == "insert":
                return dao.insert(Event.from_json(p["event"]), app_id, ch)
            if method == "insert_batch":
                return dao.insert_batch(
                    [Event.from_json(d) for d in p["events"]], app_id, ch)
            if method == "get":
                e = dao.get(p["eventId"], app_id, ch)
                return e.to_json() if e else None
            if method == "delete":
                return dao.delete(p["eventId"], app_id, ch)
            if method == "find":
                rows = dao.find(
                    app_id=app_id, channel_id=ch,
                    start_time=from_ms(p.get("startTime")),
                    until_time=from_ms(p.get("untilTime")),
                    entity_type=p.get("entityType"),
                    entity_id=p.get("entityId"),
                    event_names=p.get("eventNames"),
                    target_entity_type=opt_unfield(
                        p.get("targetEntityType")),
                    target_entity_id=opt_unfield(p.get("targetEntityId")),
                    limit=p.get("limit"), reversed=p.get("reversed", False))
                return [e.to_json() for e in rows]
            if method == "find_columns":
                return dao.find_columns(
                    app_id=app_id, channel_id=ch,
                    start_time=from_ms(p.get("startTime")),
                    until_time=from_ms(p.get("untilTime")),
                    entity_type=p.get("entityType"),
                    event_names=p.get("eventNames"),
                    target_entity_type=opt_unfield(
                        p.get("targetEntityType")),
                    property_fields=p.get("propertyFields") or ())
            if method == "aggregate_properties":
                out = dao.aggregate_properties(
                    app_id=app_id, channel_id=ch,
                    entity_type=p["entityType"],
                    start_time=from_ms(p.get("startTime")),
                    until_time=from_ms(p.get("untilTime")),
                    required=p.get("required"))
                return {k: {"fields": v.to_dict(),
                            "firstUpdated": ms(v.first_updated),
                            "lastUpdated": ms(v.last_updated)}
                        for k, v in out.items()}
        raise ValueError(f"unknown rpc {kind}.{method}")

    # optional shared-secret auth (the PostgreSQL-password analog):
    # set PIO_STORAGE_SERVER_KEY on the daemon; clients send it via the
    # X-PIO-Storage-Key header (remote driver: source config KEY=...)
    import os as _os
    auth_key = _os.environ.get("PIO_STORAGE_SERVER_KEY")

    @app.get("/")
    def index():
        return {"status": "alive", "service": "pio-storage-server"}

    @app.post("/s/{kind}/{method}")
    async def rpc(kind: str, method: str, request: Request):
        if auth_key and request.headers.get("X-PIO-Storage-Key") != auth_key:
            return JSONResponse({"message": "invalid storage key"},
                                status_code=401)
        try:
            payload = await request.json()
        except Exception:
            return JSONResponse({"message": "invalid JSON"},
                                status_code=400)
        try:
            # DB work runs in the threadpool so slow queries don't
            # stall the event loop for other clients
            from starlette.concurrency import run_in_threadpool
            return {"r": await run_in_threadpool(handle, kind, method,
                                                 payload)}
        except ValueError as e:
            return JSONResponse({"message": str(e)}, status_code=404)
        except Exception as e:  # noqa: BLE001
            logger.exception("storage rpc failed")
            return JSONResponse({"message": str(e)}, status_code=500)

    return app


def run(host: str = "0.0.0.0", port: int = 7072):
    import uvicorn
    uvicorn.run(create_app(), host=host, port=port, log_level="info")


if __name__ == "__main__":
    run()
