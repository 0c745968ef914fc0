"""Admin REST API (port 7071, experimental — like the reference's).

Parity with tools/.../admin/AdminAPI.scala:50-120 + CommandClient.scala:
- GET  /                       → server status
- GET  /status                 → deep storage health (pio status)
- GET  /cmd/app                → list apps
- POST /cmd/app                → create app {"name": ..., ["description"]}
- GET  /cmd/app/{name}         → app detail (keys + channels)
- DELETE /cmd/app/{name}       → delete app + keys + events
- DELETE /cmd/app/{name}/data  → wipe the app's event data
- POST /cmd/app/{name}/accesskey          → new access key
- DELETE /cmd/accesskey/{key}             → revoke a key
- POST /cmd/app/{name}/channel            → new channel {"name": ...}
- DELETE /cmd/app/{name}/channel/{chname} → delete channel + its events
- GET  /cmd/engineinstances    → list training runs
"""

from __future__ import annotations

import secrets

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse

from predictionio_amd.data import storage
from predictionio_amd.data.storage.base import AccessKey, App, Channel


def create_app() -> FastAPI:
    app = FastAPI(title="PredictionIO-AMD Admin API")

    @app.get("/")
    def index():
        return {"status": "alive"}

    @app.get("/status")
    def deep_status():
        """pio status semantics: verify every configured repository
        (Storage.verifyAllDataObjects, Storage.scala:372-394)."""
        try:
            storage.verify_all_data_objects()
            return {"status": "ok", "storage": "verified"}
        except Exception as e:  # noqa: BLE001
            return JSONResponse({"status": "error", "message": str(e)},
                                status_code=500)

    @app.get("/cmd/app/{name}")
    def app_detail(name: str):
        a = storage.get_meta_data_apps().get_by_name(name)
        if a is None:
            return JSONResponse({"message": f"App {name} does not exist."},
                                status_code=404)
        keys = storage.get_meta_data_access_keys().get_by_app_id(a.id)
        chans = storage.get_meta_data_channels().get_by_app_id(a.id)
        return {"name": a.name, "id": a.id, "description": a.description,
                "accessKeys": [{"key": k.key, "events": k.events}
                               for k in keys],
                "channels": [{"id": c.id, "name": c.name} for c in chans]}

    @app.post("/cmd/app/{name}/accesskey")
    async def new_key(name: str, request: Request):
        a = storage.get_meta_data_apps().get_by_name(name)
        if a is None:
            return JSONResponse({"message": f"App {name} does not exist."},
                                status_code=404)
        try:
            body = await request.json()
        except Exception:
            body = {}
        key = secrets.token_urlsafe(48)
        storage.get_meta_data_access_keys().insert(
            AccessKey(key=key, appid=a.id,
                      events=body.get("events") or []))
        return {"accessKey": key, "appId": a.id}

    @app.delete("/cmd/accesskey/{key}")
    def delete_key(key: str):
        if not storage.get_meta_data_access_keys().delete(key):
            return JSONResponse({"message": "key not found"},
                                status_code=404)
        return {"message": "Access key deleted."}

    @app.post("/cmd/app/{name}/channel")
    async def new_channel(name: str, request: Request):
        a = storage.get_meta_data_apps().get_by_name(name)
        if a is None:
            return JSONResponse({"message": f"App {name} does not exist."},
                                status_code=404)
        body = await request.json()
        chname = body.get("name", "")
        if not Channel.is_valid_name(chname):
            return JSONResponse(
                {"message": "Channel name must match [a-zA-Z0-9-]{1,16}"},
                status_code=400)
        cid = storage.get_meta_data_channels().insert(
            Channel(0, chname, a.id))
        storage.get_l_events().init(a.id, cid)
        return {"channel": chname, "id": cid}

    @app.delete("/cmd/app/{name}/channel/{chname}")
    def delete_channel(name: str, chname: str):
        a = storage.get_meta_data_apps().get_by_name(name)
        if a is None:
            return JSONResponse({"message": f"App {name} does not exist."},
                                status_code=404)
        for c in storage.get_meta_data_channels().get_by_app_id(a.id):
            if c.name == chname:
                storage.get_l_events().remove(a.id, c.id)
                storage.get_meta_data_channels().delete(c.id)
                return {"message": f"Channel {chname} deleted."}
        return JSONResponse({"message": f"Channel {chname} not found."},
                            status_code=404)

    @app.get("/cmd/engineinstances")
    def list_engine_instances():
        insts = storage.get_meta_data_engine_instances().get_all()
        return {"engineInstances": [
            {"id": i.id, "status": i.status,
             "engineFactory": i.engine_factory,
             "engineVariant": i.engine_variant,
             "startTime": str(i.start_time), "endTime": str(i.end_time)}
            for i in insts]}

    @app.get("/cmd/app")
    def list_apps():
        apps = storage.get_meta_data_apps().get_all()
        keys = storage.get_meta_data_access_keys()
        return {"apps": [
            {"name": a.name, "id": a.id,
             "description": a.description,
             "accessKeys": [k.key for k in keys.get_by_app_id(a.id)]}
            for a in apps]}

    @app.post("/cmd/app")
    async def new_app(request: Request):
        body = await request.json()
        name = body.get("name")
        if not name:
            return JSONResponse({"message": "name is required"},
                                status_code=400)
        apps = storage.get_meta_data_apps()
        if apps.get_by_name(name):
            return JSONResponse(
                {"message": f"App {name} already exists."}, status_code=409)
        app_id = apps.insert(App(id=body.get("id", 0), name=name,
                                 description=body.get("description")))
        storage.get_l_events().init(app_id)
        key = secrets.token_urlsafe(48)
        storage.get_meta_data_access_keys().insert(
            AccessKey(key=key, appid=app_id, events=[]))
        return {"name": name, "id": app_id, "accessKey": key}

    @app.delete("/cmd/app/{name}")
    def delete_app(name: str):
        apps = storage.get_meta_data_apps()
        a = apps.get_by_name(name)
        if a is None:
            return JSONResponse({"message": f"App {name} does not exist."},
                                status_code=404)
        for c in storage.get_meta_data_channels().get_by_app_id(a.id):
            storage.get_l_events().remove(a.id, c.id)
            storage.get_meta_data_channels().delete(c.id)
        storage.get_l_events().remove(a.id)
        for k in storage.get_meta_data_access_keys().get_by_app_id(a.id):
            storage.get_meta_data_access_keys().delete(k.key)
        apps.delete(a.id)
        return {"message": f"App {name} deleted."}

    @app.delete("/cmd/app/{name}/data")
    def delete_app_data(name: str):
        a = storage.get_meta_data_apps().get_by_name(name)
        if a is None:
            return JSONResponse({"message": f"App {name} does not exist."},
                                status_code=404)
        storage.get_l_events().remove(a.id)
        storage.get_l_events().init(a.id)
        return {"message": f"Data of app {name} deleted."}

    return app


def run(host: str = "localhost", port: int = 7071) -> None:
    import uvicorn
    uvicorn.run(create_app(), host=host, port=port, log_level="info")
