"""Admin REST API (port 7071, experimental — like the reference's).

Parity with tools/.../admin/AdminAPI.scala:50-120 + CommandClient.scala:
- GET  /                       → server status
- GET  /cmd/app                → list apps
- POST /cmd/app                → create app {"name": ..., ["description"]}
- DELETE /cmd/app/{name}       → delete app + keys + events
- DELETE /cmd/app/{name}/data  → wipe the app's event data
"""

from __future__ import annotations

import secrets

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse

from predictionio_amd.data import storage
from predictionio_amd.data.storage.base import AccessKey, App


def create_app() -> FastAPI:
    app = FastAPI(title="PredictionIO-AMD Admin API")

    @app.get("/")
    def index():
        return {"status": "alive"}

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
