"""SDK-compatible HTTP clients.

Parity with the official PredictionIO Python SDK surface (the reference
framework's REST contract is consumed through `predictionio.EventClient` /
`predictionio.EngineClient`): same constructor shapes and method names so
SDK-based application code ports by changing only the import. Synchronous
(requests); the a* variants exist for signature compatibility and return
the completed result immediately.
"""

from __future__ import annotations

import datetime
from typing import Any, Dict, List, Optional

import requests


def _fmt_time(t) -> Optional[str]:
    if t is None:
        return None
    if isinstance(t, datetime.datetime):
        if t.tzinfo is None:
            t = t.replace(tzinfo=datetime.timezone.utc)
        return t.isoformat()
    return str(t)


class PredictionIOAPIError(Exception):
    pass


class NotCreatedError(PredictionIOAPIError):
    pass


class NotFoundError(PredictionIOAPIError):
    pass


class BaseClient:
    def __init__(self, url: str, threads: int = 1, qsize: int = 0,
                 timeout: float = 5.0):
        self.url = url.rstrip("/")
        self.timeout = timeout
        self._s = requests.Session()

    def close(self):
        self._s.close()


class EventClient(BaseClient):
    """Event-server client (SDK: predictionio.EventClient)."""

    def __init__(self, access_key: str, url: str = "http://localhost:7070",
                 threads: int = 1, qsize: int = 0, timeout: float = 5.0,
                 channel: Optional[str] = None):
        super().__init__(url, threads, qsize, timeout)
        self.access_key = access_key
        self.channel = channel

    def _params(self) -> Dict[str, str]:
        p = {"accessKey": self.access_key}
        if self.channel:
            p["channel"] = self.channel
        return p

    # ---- generic events

    def create_event(self, event: str, entity_type: str, entity_id: str,
                     target_entity_type: Optional[str] = None,
                     target_entity_id: Optional[str] = None,
                     properties: Optional[dict] = None,
                     event_time=None) -> dict:
        body: Dict[str, Any] = {
            "event": event, "entityType": entity_type,
            "entityId": entity_id,
        }
        if target_entity_type is not None:
            body["targetEntityType"] = target_entity_type
        if target_entity_id is not None:
            body["targetEntityId"] = target_entity_id
        if properties is not None:
            body["properties"] = properties
        t = _fmt_time(event_time)
        if t:
            body["eventTime"] = t
        r = self._s.post(f"{self.url}/events.json", json=body,
                         params=self._params(), timeout=self.timeout)
        if r.status_code != 201:
            raise NotCreatedError(f"event not created: {r.text}")
        return r.json()

    acreate_event = create_event

    def get_event(self, event_id: str) -> dict:
        r = self._s.get(f"{self.url}/events/{event_id}.json",
                        params=self._params(), timeout=self.timeout)
        if r.status_code != 200:
            raise NotFoundError(r.text)
        return r.json()

    aget_event = get_event

    def delete_event(self, event_id: str) -> dict:
        r = self._s.delete(f"{self.url}/events/{event_id}.json",
                           params=self._params(), timeout=self.timeout)
        if r.status_code != 200:
            raise NotFoundError(r.text)
        return r.json()

    adelete_event = delete_event

    def get_events(self, **filters) -> List[dict]:
        params = dict(self._params())
        params.update({k: v for k, v in filters.items() if v is not None})
        r = self._s.get(f"{self.url}/events.json", params=params,
                        timeout=self.timeout)
        if r.status_code == 404:
            return []
        r.raise_for_status()
        return r.json()

    # ---- user convenience (SDK aset_user/aunset_user/adelete_user)

    def set_user(self, uid: str, properties: Optional[dict] = None,
                 event_time=None) -> dict:
        return self.create_event("$set", "user", uid,
                                 properties=properties or {},
                                 event_time=event_time)

    def unset_user(self, uid: str, properties: dict,
                   event_time=None) -> dict:
        return self.create_event("$unset", "user", uid,
                                 properties=properties,
                                 event_time=event_time)

    def delete_user(self, uid: str, event_time=None) -> dict:
        return self.create_event("$delete", "user", uid,
                                 event_time=event_time)

    aset_user = set_user
    aunset_user = unset_user
    adelete_user = delete_user

    # ---- item convenience

    def set_item(self, iid: str, properties: Optional[dict] = None,
                 event_time=None) -> dict:
        return self.create_event("$set", "item", iid,
                                 properties=properties or {},
                                 event_time=event_time)

    def unset_item(self, iid: str, properties: dict,
                   event_time=None) -> dict:
        return self.create_event("$unset", "item", iid,
                                 properties=properties,
                                 event_time=event_time)

    def delete_item(self, iid: str, event_time=None) -> dict:
        return self.create_event("$delete", "item", iid,
                                 event_time=event_time)

    aset_item = set_item
    aunset_item = unset_item
    adelete_item = delete_item

    # ---- user-to-item action (SDK arecord_user_action_on_item)

    def record_user_action_on_item(self, action: str, uid: str, iid: str,
                                   properties: Optional[dict] = None,
                                   event_time=None) -> dict:
        return self.create_event(action, "user", uid, "item", iid,
                                 properties=properties,
                                 event_time=event_time)

    arecord_user_action_on_item = record_user_action_on_item


class EngineClient(BaseClient):
    """Engine-server client (SDK: predictionio.EngineClient)."""

    def __init__(self, url: str = "http://localhost:8000",
                 threads: int = 1, qsize: int = 0, timeout: float = 5.0):
        super().__init__(url, threads, qsize, timeout)

    def send_query(self, data: dict) -> dict:
        r = self._s.post(f"{self.url}/queries.json", json=data,
                         timeout=self.timeout)
        if r.status_code != 200:
            raise PredictionIOAPIError(
                f"query failed ({r.status_code}): {r.text}")
        return r.json()

    asend_query = send_query
