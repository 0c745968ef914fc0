"""Event Server — REST ingest API.

Parity with the reference Event Server (data/.../api/EventServer.scala):
- accessKey auth via query param or HTTP basic credentials, with channel
  resolution (EventServer.scala:92-130)
- GET /                             → {"status": "alive"} (:148-152)
- POST /events.json                 → 201 {"eventId": ...} (:241-273);
  403 when the key restricts event names (:267-268)
- GET /events.json + 9 filter dims  → 200 array | 404 (:274-339)
- GET/DELETE /events/<id>.json      (:207-240)
- POST /batch/events.json           ≤ 50 events, per-event statuses
  (:66, :340-420)
- GET /stats.json                   (:421-441; --stats flag)
- POST/GET /webhooks/<c>.json|.form (:442-523; api/Webhooks.scala:32-138)
- input blocker / sniffer plugins   (EventServerPlugin.scala:22-34)

The reference is Akka-HTTP on port 7070; here FastAPI+uvicorn, same port,
same wire format — SDK clients work unchanged.
"""

from __future__ import annotations

import base64
import logging
from dataclasses import dataclass
from typing import Any, Dict, List, Optional

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse

from predictionio_amd.data import storage
from predictionio_amd.data.events import Event, parse_time, validate_event
from predictionio_amd.server.stats import Stats
from predictionio_amd.server.webhooks import (
    ConnectorException, form_connectors, json_connectors,
)

logger = logging.getLogger(__name__)

MAX_BATCH_SIZE = 50  # EventServer.scala:66


class EventServerPlugin:
    """Input blocker / sniffer SPI (EventServerPlugin.scala:22-34)."""

    inputblocker = "inputblocker"
    inputsniffer = "inputsniffer"

    plugin_name = "plugin"
    plugin_description = ""
    plugin_type = inputsniffer

    def process(self, event_info: dict) -> None:
        """Blockers raise to reject the event; sniffers observe."""

    def handle_rest(self, arguments: dict) -> Any:
        return {}


@dataclass
class AuthData:
    """Resolved access key (EventServer.scala AuthData)."""
    app_id: int
    channel_id: Optional[int]
    events: List[str]


def _unauthorized(msg: str = "Invalid accessKey.") -> JSONResponse:
    return JSONResponse({"message": msg}, status_code=401)


def create_app(stats_on: bool = False,
               plugins: Optional[List[EventServerPlugin]] = None) -> FastAPI:
    app = FastAPI(title="PredictionIO-AMD Event Server")
    stats = Stats()
    plugins = plugins or []
    blockers = [p for p in plugins
                if p.plugin_type == EventServerPlugin.inputblocker]
    sniffers = [p for p in plugins
                if p.plugin_type == EventServerPlugin.inputsniffer]

    levents = storage.get_l_events()
    access_keys = storage.get_meta_data_access_keys()
    channels = storage.get_meta_data_channels()

    def authenticate(request: Request) -> Optional[AuthData]:
        """Key from ?accessKey= or HTTP basic username
        (EventServer.scala:92-130)."""
        key = request.query_params.get("accessKey")
        if not key:
            auth = request.headers.get("authorization", "")
            if auth.lower().startswith("basic "):
                try:
                    decoded = base64.b64decode(auth[6:]).decode()
                    key = decoded.split(":", 1)[0]
                except Exception:
                    key = None
        if not key:
            return None
        ak = access_keys.get(key)
        if ak is None:
            return None
        channel_name = request.query_params.get("channel")
        channel_id = None
        if channel_name:
            ch = [c for c in channels.get_by_app_id(ak.appid)
                  if c.name == channel_name]
            if not ch:
                return None
            channel_id = ch[0].id
        return AuthData(app_id=ak.appid, channel_id=channel_id,
                        events=ak.events)

    def _notify(event: Event, auth: AuthData) -> Optional[str]:
        """Run blockers (returning rejection message) then sniffers."""
        info = {"appId": auth.app_id, "channelId": auth.channel_id,
                "event": event}
        for b in blockers:
            try:
                b.process(info)
            except Exception as e:  # blocker rejects
                return str(e)
        for s in sniffers:
            try:
                s.process(info)
            except Exception:
                logger.exception("sniffer plugin failed")
        return None

    def _insert_one(data: Dict[str, Any], auth: AuthData) -> tuple:
        """→ (status, body) for one event submission."""
        try:
            event = Event.from_json(data)
            validate_event(event)
        except Exception as e:
            return 400, {"message": str(e)}
        if auth.events and event.event not in auth.events:
            return 403, {"message": f"{event.event} events are not allowed"}
        blocked = _notify(event, auth)
        if blocked is not None:
            return 403, {"message": blocked}
        eid = levents.insert(event, auth.app_id, auth.channel_id)
        return 201, {"eventId": eid}

    # ------------------------------------------------------------ routes

    @app.get("/")
    def index():
        return {"status": "alive"}

    @app.get("/plugins.json")
    def list_plugins():
        return {"plugins": {
            p.plugin_name: {"name": p.plugin_name,
                            "description": p.plugin_description,
                            "class": type(p).__name__}
            for p in plugins
        }}

    @app.get("/plugins/{name}/{rest:path}")
    async def plugin_rest(name: str, rest: str, request: Request):
        """REST passthrough to a named plugin
        (EventServer.scala:183-192 HandleREST)."""
        for p in plugins:
            if p.plugin_name == name:
                return p.handle_rest({"path": rest,
                                      "query": dict(request.query_params)})
        return JSONResponse({"message": f"plugin {name} not found"},
                            status_code=404)

    @app.post("/events.json")
    async def post_event(request: Request):
        auth = authenticate(request)
        if auth is None:
            return _unauthorized()
        try:
            data = await request.json()
        except Exception:
            stats_on and stats.bookkeeping(0, 400)
            return JSONResponse({"message": "invalid JSON"}, status_code=400)
        status, body = _insert_one(data, auth)
        if stats_on:
            ev = None
            if status == 201:
                try:
                    ev = Event.from_json(data)
                except Exception:
                    pass
            stats.bookkeeping(auth.app_id, status, ev)
        return JSONResponse(body, status_code=status)

    @app.get("/events.json")
    def get_events(request: Request):
        auth = authenticate(request)
        if auth is None:
            return _unauthorized()
        q = request.query_params
        reversed_ = q.get("reversed") == "true"
        entity_type = q.get("entityType")
        entity_id = q.get("entityId")
        if reversed_ and not (entity_type and entity_id):
            return JSONResponse(
                {"message": "the parameter reversed can only be used with "
                            "both entityType and entityId specified."},
                status_code=400)
        try:
            start_time = parse_time(q["startTime"]) if "startTime" in q \
                else None
            until_time = parse_time(q["untilTime"]) if "untilTime" in q \
                else None
        except Exception as e:
            return JSONResponse({"message": str(e)}, status_code=400)
        limit = int(q.get("limit", 20))
        events = levents.find(
            auth.app_id, auth.channel_id,
            start_time=start_time, until_time=until_time,
            entity_type=entity_type, entity_id=entity_id,
            event_names=[q["event"]] if "event" in q else None,
            target_entity_type=(q["targetEntityType"]
                                if "targetEntityType" in q
                                else storage.base.UNSET),
            target_entity_id=(q["targetEntityId"]
                              if "targetEntityId" in q
                              else storage.base.UNSET),
            limit=limit, reversed=reversed_)
        out = [e.to_json() for e in events]
        if not out:
            return JSONResponse({"message": "Not Found"}, status_code=404)
        return out

    @app.get("/events/{event_id}.json")
    def get_event(event_id: str, request: Request):
        auth = authenticate(request)
        if auth is None:
            return _unauthorized()
        e = levents.get(event_id, auth.app_id, auth.channel_id)
        if e is None:
            return JSONResponse({"message": "Not Found"}, status_code=404)
        return e.to_json()

    @app.delete("/events/{event_id}.json")
    def delete_event(event_id: str, request: Request):
        auth = authenticate(request)
        if auth is None:
            return _unauthorized()
        if levents.delete(event_id, auth.app_id, auth.channel_id):
            return {"message": "Found"}
        return JSONResponse({"message": "Not Found"}, status_code=404)

    @app.post("/batch/events.json")
    async def batch_events(request: Request):
        auth = authenticate(request)
        if auth is None:
            return _unauthorized()
        try:
            items = await request.json()
            assert isinstance(items, list)
        except Exception:
            return JSONResponse({"message": "invalid JSON array"},
                                status_code=400)
        if len(items) > MAX_BATCH_SIZE:
            return JSONResponse(
                {"message": "Batch request must have less than or equal to "
                            f"{MAX_BATCH_SIZE} events"},
                status_code=400)
        # validate/authorize every item first, then ONE bulk DB write for
        # the valid ones (the reference batches via async futures,
        # LEvents.futureInsertBatch LEvents.scala:106-112; here the DAO's
        # insert_batch is a single executemany). Per-event statuses keep
        # their order.
        results = []
        valid: list = []  # (result_slot_index, Event)
        for i, item in enumerate(items):
            try:
                event = Event.from_json(item)
                validate_event(event)
            except Exception as e:
                results.append({"status": 400, "message": str(e)})
                continue
            if auth.events and event.event not in auth.events:
                results.append({"status": 403,
                                "message": f"{event.event} events are not "
                                           "allowed"})
                continue
            blocked = _notify(event, auth)
            if blocked is not None:
                results.append({"status": 403, "message": blocked})
                continue
            results.append(None)
            valid.append((len(results) - 1, event))
        if valid:
            try:
                ids = levents.insert_batch([e for _, e in valid],
                                           auth.app_id, auth.channel_id)
                for (slot, _), eid in zip(valid, ids):
                    results[slot] = {"status": 201, "eventId": eid}
            except Exception as e:
                for slot, _ in valid:
                    results[slot] = {"status": 500, "message": str(e)}
        if stats_on:
            for r in results:
                stats.bookkeeping(auth.app_id, r["status"])
        return results

    @app.get("/stats.json")
    def get_stats(request: Request):
        auth = authenticate(request)
        if auth is None:
            return _unauthorized()
        if not stats_on:
            return JSONResponse(
                {"message": "To see stats, launch Event Server with "
                            "--stats argument."}, status_code=404)
        return stats.get(auth.app_id)

    # ------------------------------------------------------------ webhooks

    jc = json_connectors()
    fc = form_connectors()

    @app.post("/webhooks/{path}.json")
    async def webhook_json(path: str, request: Request):
        auth = authenticate(request)
        if auth is None:
            return _unauthorized()
        conn = jc.get(path)
        if conn is None:
            return JSONResponse(
                {"message": f"webhooks connection for {path} is not "
                            "supported."}, status_code=404)
        try:
            data = await request.json()
            event_json = conn.to_event_json(data)
        except ConnectorException as e:
            return JSONResponse({"message": str(e)}, status_code=400)
        except Exception:
            return JSONResponse({"message": "invalid JSON"}, status_code=400)
        status, body = _insert_one(event_json, auth)
        return JSONResponse(body, status_code=status)

    @app.get("/webhooks/{path}.json")
    def webhook_json_get(path: str, request: Request):
        auth = authenticate(request)
        if auth is None:
            return _unauthorized()
        if path not in jc:
            return JSONResponse(
                {"message": f"webhooks connection for {path} is not "
                            "supported."}, status_code=404)
        return {}

    @app.post("/webhooks/{path}.form")
    async def webhook_form(path: str, request: Request):
        auth = authenticate(request)
        if auth is None:
            return _unauthorized()
        conn = fc.get(path)
        if conn is None:
            return JSONResponse(
                {"message": f"webhooks connection for {path} is not "
                            "supported."}, status_code=404)
        # parse application/x-www-form-urlencoded directly (no dependency
        # on python-multipart)
        from urllib.parse import parse_qsl
        body = (await request.body()).decode()
        form = dict(parse_qsl(body, keep_blank_values=True))
        try:
            event_json = conn.to_event_json(form)
        except ConnectorException as e:
            return JSONResponse({"message": str(e)}, status_code=400)
        status, body = _insert_one(event_json, auth)
        return JSONResponse(body, status_code=status)

    @app.get("/webhooks/{path}.form")
    def webhook_form_get(path: str, request: Request):
        auth = authenticate(request)
        if auth is None:
            return _unauthorized()
        if path not in fc:
            return JSONResponse(
                {"message": f"webhooks connection for {path} is not "
                            "supported."}, status_code=404)
        return {}

    return app


def run(host: str = "0.0.0.0", port: int = 7070, stats_on: bool = False,
        ssl_keyfile: str = None, ssl_certfile: str = None) -> None:
    """`pio eventserver` entry point (EventServer.scala Run.main:551-560;
    default port 7070 as in the reference). SSL per the reference's
    SSLConfiguration (common/.../SSLConfiguration.scala) via uvicorn."""
    import uvicorn
    uvicorn.run(create_app(stats_on=stats_on), host=host, port=port,
                log_level="info", ssl_keyfile=ssl_keyfile,
                ssl_certfile=ssl_certfile)
