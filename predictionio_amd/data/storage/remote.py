"""Client-server storage backend: the `remote` driver.

Plays the role of the reference's production JDBC PostgreSQL/MySQL
backend (storage/jdbc/.../JDBCLEvents.scala:55-88, JDBCModels.scala:55):
a long-running storage DAEMON (predictionio_amd.server.storageserver,
`pio storageserver`, port 7072) owns the database, and every framework
process — Event Server, trainer, query server, dashboard — talks to it
over HTTP with this driver. That gives the reference's operational
class: one server process handling concurrent multi-process access,
instead of N processes hammering one SQLite file.

Wire format: JSON RPC-style POST /s/{kind}/{method}; events use the
reference event wire schema (EventJson4sSupport), model blobs are
base64, datetimes are epoch milliseconds. The same DAO contract tests
run over sqlite / memory / remote (tests/test_storage_remote.py).

Registry config:
    PIO_STORAGE_SOURCES_<NAME>_TYPE=remote
    PIO_STORAGE_SOURCES_<NAME>_URL=http://127.0.0.1:7072
"""

from __future__ import annotations

import base64
from datetime import datetime, timezone
from typing import Any, Dict, Iterable, List, Optional, Sequence

from predictionio_amd.data.events import DataMap, Event, PropertyMap
from predictionio_amd.data.storage import base
from predictionio_amd.data.storage.base import (
    UNSET, AccessKey, App, Channel, EngineInstance, EvaluationInstance,
    Model,
)

# ------------------------------------------------------------- marshalling


def ms(dt: Optional[datetime]) -> Optional[int]:
    return None if dt is None else int(dt.timestamp() * 1000)


def from_ms(v) -> Optional[datetime]:
    return (None if v is None
            else datetime.fromtimestamp(v / 1000.0, tz=timezone.utc))


def ei_to_dict(i: EngineInstance) -> Dict[str, Any]:
    return {
        "id": i.id, "status": i.status, "startTime": ms(i.start_time),
        "endTime": ms(i.end_time), "engineId": i.engine_id,
        "engineVersion": i.engine_version,
        "engineVariant": i.engine_variant,
        "engineFactory": i.engine_factory, "batch": i.batch, "env": i.env,
        "runtimeConf": i.runtime_conf,
        "dataSourceParams": i.data_source_params,
        "preparatorParams": i.preparator_params,
        "algorithmsParams": i.algorithms_params,
        "servingParams": i.serving_params,
    }


def ei_from_dict(d: Dict[str, Any]) -> EngineInstance:
    return EngineInstance(
        id=d["id"], status=d["status"],
        start_time=from_ms(d["startTime"]), end_time=from_ms(d["endTime"]),
        engine_id=d["engineId"], engine_version=d["engineVersion"],
        engine_variant=d["engineVariant"],
        engine_factory=d["engineFactory"], batch=d.get("batch", ""),
        env=d.get("env") or {}, runtime_conf=d.get("runtimeConf") or {},
        data_source_params=d.get("dataSourceParams", ""),
        preparator_params=d.get("preparatorParams", ""),
        algorithms_params=d.get("algorithmsParams", ""),
        serving_params=d.get("servingParams", ""))


def evi_to_dict(i: EvaluationInstance) -> Dict[str, Any]:
    return {
        "id": i.id, "status": i.status, "startTime": ms(i.start_time),
        "endTime": ms(i.end_time), "evaluationClass": i.evaluation_class,
        "engineParamsGeneratorClass": i.engine_params_generator_class,
        "batch": i.batch, "env": i.env,
        "evaluatorResults": i.evaluator_results,
        "evaluatorResultsHTML": i.evaluator_results_html,
        "evaluatorResultsJSON": i.evaluator_results_json,
    }


def evi_from_dict(d: Dict[str, Any]) -> EvaluationInstance:
    return EvaluationInstance(
        id=d["id"], status=d["status"],
        start_time=from_ms(d["startTime"]), end_time=from_ms(d["endTime"]),
        evaluation_class=d.get("evaluationClass", ""),
        engine_params_generator_class=d.get(
            "engineParamsGeneratorClass", ""),
        batch=d.get("batch", ""), env=d.get("env") or {},
        evaluator_results=d.get("evaluatorResults", ""),
        evaluator_results_html=d.get("evaluatorResultsHTML", ""),
        evaluator_results_json=d.get("evaluatorResultsJSON", ""))


def opt_field(v: Any) -> Dict[str, Any]:
    """UNSET/None/value tri-state → wire: omit / {"null": true} / value."""
    if v is UNSET:
        return {}
    if v is None:
        return {"null": True}
    return {"value": v}


def opt_unfield(d: Optional[Dict[str, Any]]):
    if not d:
        return UNSET
    if d.get("null"):
        return None
    return d.get("value")


# ------------------------------------------------------------------ client


class RemoteClient:
    """One storage source = one storage-server URL."""

    def __init__(self, url: str, timeout: float = 30.0,
                 key: str = None):
        import httpx
        self.url = url.rstrip("/")
        headers = {"X-PIO-Storage-Key": key} if key else None
        self._http = httpx.Client(base_url=self.url, timeout=timeout,
                                  headers=headers)

    def call(self, kind: str, method: str, payload: Dict[str, Any]) -> Any:
        r = self._http.post(f"/s/{kind}/{method}", json=payload)
        if r.status_code != 200:
            raise RuntimeError(
                f"storage server {kind}.{method} -> {r.status_code}: "
                f"{r.text[:500]}")
        return r.json()["r"]

    def close(self):
        self._http.close()


class RemoteApps(base.Apps):
    def __init__(self, client: RemoteClient):
        self.c = client

    def insert(self, app: App) -> Optional[int]:
        return self.c.call("apps", "insert", {
            "id": app.id, "name": app.name,
            "description": app.description})

    def get(self, app_id: int) -> Optional[App]:
        d = self.c.call("apps", "get", {"id": app_id})
        return App(**d) if d else None

    def get_by_name(self, name: str) -> Optional[App]:
        d = self.c.call("apps", "get_by_name", {"name": name})
        return App(**d) if d else None

    def get_all(self) -> List[App]:
        return [App(**d) for d in self.c.call("apps", "get_all", {})]

    def update(self, app: App) -> bool:
        return self.c.call("apps", "update", {
            "id": app.id, "name": app.name,
            "description": app.description})

    def delete(self, app_id: int) -> bool:
        return self.c.call("apps", "delete", {"id": app_id})


class RemoteAccessKeys(base.AccessKeys):
    def __init__(self, client: RemoteClient):
        self.c = client

    def insert(self, k: AccessKey) -> Optional[str]:
        return self.c.call("accesskeys", "insert", {
            "key": k.key, "appid": k.appid, "events": k.events})

    def get(self, key: str) -> Optional[AccessKey]:
        d = self.c.call("accesskeys", "get", {"key": key})
        return AccessKey(**d) if d else None

    def get_all(self) -> List[AccessKey]:
        return [AccessKey(**d)
                for d in self.c.call("accesskeys", "get_all", {})]

    def get_by_app_id(self, app_id: int) -> List[AccessKey]:
        return [AccessKey(**d) for d in self.c.call(
            "accesskeys", "get_by_app_id", {"appid": app_id})]

    def update(self, k: AccessKey) -> bool:
        return self.c.call("accesskeys", "update", {
            "key": k.key, "appid": k.appid, "events": k.events})

    def delete(self, key: str) -> bool:
        return self.c.call("accesskeys", "delete", {"key": key})


class RemoteChannels(base.Channels):
    def __init__(self, client: RemoteClient):
        self.c = client

    def insert(self, ch: Channel) -> Optional[int]:
        return self.c.call("channels", "insert", {
            "id": ch.id, "name": ch.name, "appid": ch.appid})

    def get(self, channel_id: int) -> Optional[Channel]:
        d = self.c.call("channels", "get", {"id": channel_id})
        return Channel(**d) if d else None

    def get_by_app_id(self, app_id: int) -> List[Channel]:
        return [Channel(**d) for d in self.c.call(
            "channels", "get_by_app_id", {"appid": app_id})]

    def delete(self, channel_id: int) -> bool:
        return self.c.call("channels", "delete", {"id": channel_id})


class RemoteEngineInstances(base.EngineInstances):
    def __init__(self, client: RemoteClient):
        self.c = client

    def insert(self, i: EngineInstance) -> str:
        return self.c.call("engineinstances", "insert", ei_to_dict(i))

    def get(self, iid: str) -> Optional[EngineInstance]:
        d = self.c.call("engineinstances", "get", {"id": iid})
        return ei_from_dict(d) if d else None

    def get_all(self) -> List[EngineInstance]:
        return [ei_from_dict(d)
                for d in self.c.call("engineinstances", "get_all", {})]

    def get_completed(self, engine_id, engine_version, engine_variant):
        return [ei_from_dict(d) for d in self.c.call(
            "engineinstances", "get_completed",
            {"engineId": engine_id, "engineVersion": engine_version,
             "engineVariant": engine_variant})]

    def get_latest_completed(self, engine_id, engine_version,
                             engine_variant):
        d = self.c.call("engineinstances", "get_latest_completed",
                        {"engineId": engine_id,
                         "engineVersion": engine_version,
                         "engineVariant": engine_variant})
        return ei_from_dict(d) if d else None

    def get_latest_completed_by_factory(self, engine_factory,
                                        engine_variant=None):
        d = self.c.call("engineinstances",
                        "get_latest_completed_by_factory",
                        {"engineFactory": engine_factory,
                         "engineVariant": engine_variant})
        return ei_from_dict(d) if d else None

    def update(self, i: EngineInstance) -> bool:
        return self.c.call("engineinstances", "update", ei_to_dict(i))

    def delete(self, iid: str) -> bool:
        return self.c.call("engineinstances", "delete", {"id": iid})


class RemoteEvaluationInstances(base.EvaluationInstances):
    def __init__(self, client: RemoteClient):
        self.c = client

    def insert(self, i: EvaluationInstance) -> str:
        return self.c.call("evaluationinstances", "insert", evi_to_dict(i))

    def get(self, iid: str) -> Optional[EvaluationInstance]:
        d = self.c.call("evaluationinstances", "get", {"id": iid})
        return evi_from_dict(d) if d else None

    def get_all(self) -> List[EvaluationInstance]:
        return [evi_from_dict(d) for d in self.c.call(
            "evaluationinstances", "get_all", {})]

    def get_completed(self) -> List[EvaluationInstance]:
        return [evi_from_dict(d) for d in self.c.call(
            "evaluationinstances", "get_completed", {})]

    def update(self, i: EvaluationInstance) -> bool:
        return self.c.call("evaluationinstances", "update", evi_to_dict(i))

    def delete(self, iid: str) -> bool:
        return self.c.call("evaluationinstances", "delete", {"id": iid})


class RemoteModels(base.Models):
    def __init__(self, client: RemoteClient):
        self.c = client

    def insert(self, m: Model) -> None:
        self.c.call("models", "insert", {
            "id": m.id,
            "models": base64.b64encode(m.models).decode("ascii")})

    def get(self, mid: str) -> Optional[Model]:
        d = self.c.call("models", "get", {"id": mid})
        if not d:
            return None
        return Model(id=d["id"], models=base64.b64decode(d["models"]))

    def delete(self, mid: str) -> bool:
        return self.c.call("models", "delete", {"id": mid})


class RemoteLEvents(base.LEvents):
    def __init__(self, client: RemoteClient):
        self.c = client

    def init(self, app_id: int, channel_id: Optional[int] = None) -> bool:
        return self.c.call("levents", "init",
                           {"appId": app_id, "channelId": channel_id})

    def remove(self, app_id: int, channel_id: Optional[int] = None) -> bool:
        return self.c.call("levents", "remove",
                           {"appId": app_id, "channelId": channel_id})

    def insert(self, event: Event, app_id: int,
               channel_id: Optional[int] = None) -> str:
        return self.c.call("levents", "insert", {
            "appId": app_id, "channelId": channel_id,
            "event": event.to_json()})

    def insert_batch(self, events: Sequence[Event], app_id: int,
                     channel_id: Optional[int] = None) -> List[str]:
        return self.c.call("levents", "insert_batch", {
            "appId": app_id, "channelId": channel_id,
            "events": [e.to_json() for e in events]})

    def get(self, event_id: str, app_id: int,
            channel_id: Optional[int] = None) -> Optional[Event]:
        d = self.c.call("levents", "get", {
            "appId": app_id, "channelId": channel_id, "eventId": event_id})
        return Event.from_json(d) if d else None

    def delete(self, event_id: str, app_id: int,
               channel_id: Optional[int] = None) -> bool:
        return self.c.call("levents", "delete", {
            "appId": app_id, "channelId": channel_id, "eventId": event_id})

    def find(self, app_id: int, channel_id: Optional[int] = None,
             start_time=None, until_time=None,
             entity_type: Optional[str] = None,
             entity_id: Optional[str] = None,
             event_names: Optional[List[str]] = None,
             target_entity_type: Any = UNSET,
             target_entity_id: Any = UNSET,
             limit: Optional[int] = None,
             reversed: bool = False) -> Iterable[Event]:
        rows = self.c.call("levents", "find", {
            "appId": app_id, "channelId": channel_id,
            "startTime": ms(start_time), "untilTime": ms(until_time),
            "entityType": entity_type, "entityId": entity_id,
            "eventNames": event_names,
            "targetEntityType": opt_field(target_entity_type),
            "targetEntityId": opt_field(target_entity_id),
            "limit": limit, "reversed": reversed})
        return (Event.from_json(d) for d in rows)

    def find_columns(self, app_id: int, channel_id: Optional[int] = None,
                     start_time=None, until_time=None,
                     entity_type: Optional[str] = None,
                     event_names: Optional[List[str]] = None,
                     target_entity_type: Any = UNSET,
                     property_fields: Sequence[str] = ()):
        return self.c.call("levents", "find_columns", {
            "appId": app_id, "channelId": channel_id,
            "startTime": ms(start_time), "untilTime": ms(until_time),
            "entityType": entity_type, "eventNames": event_names,
            "targetEntityType": opt_field(target_entity_type),
            "propertyFields": list(property_fields)})

    def aggregate_properties(self, app_id: int, entity_type: str,
                             channel_id: Optional[int] = None,
                             start_time=None, until_time=None,
                             required: Optional[List[str]] = None
                             ) -> Dict[str, PropertyMap]:
        out = self.c.call("levents", "aggregate_properties", {
            "appId": app_id, "channelId": channel_id,
            "entityType": entity_type, "startTime": ms(start_time),
            "untilTime": ms(until_time), "required": required})
        return {
            k: PropertyMap(DataMap(v["fields"]).to_dict(),
                           from_ms(v["firstUpdated"]),
                           from_ms(v["lastUpdated"]))
            for k, v in out.items()}
