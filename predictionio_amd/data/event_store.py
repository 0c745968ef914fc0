"""Event-store facades used by engine templates.

Parity:
- PEventStore.find/aggregateProperties (data/.../store/PEventStore.scala:59-119)
- LEventStore.findByEntity/find (data/.../store/LEventStore.scala:76-265)
- appName→(appId, channelId) resolution (data/.../store/Common.scala:60)

In the reference, PEventStore returns Spark RDDs for training reads and
LEventStore does blocking local reads at serving time with a timeout. Here
both resolve app/channel names and read through the LEvents DAO; the
training path (`find`) returns a list that the Preparator turns into device
tensors, and the serving path (`find_by_entity`) enforces the reference's
timeout convention (default 10 s; templates use 200 ms).
"""

from __future__ import annotations

import concurrent.futures
import logging
import threading
from datetime import datetime
from typing import Any, Dict, List, Optional

from predictionio_amd.data import storage
from predictionio_amd.data.events import Event, PropertyMap
from predictionio_amd.data.storage.base import UNSET

logger = logging.getLogger(__name__)

_POOL_WORKERS = 8
_pool = concurrent.futures.ThreadPoolExecutor(max_workers=_POOL_WORKERS)
# in-flight serving lookups (submitted, not yet finished). Timed-out
# queries are cancelled if still queued; ones already running keep a
# worker busy until the DB answers, so this gauge is the saturation
# signal (abandoned queries can pile up under a slow store).
_inflight = 0
_inflight_lock = threading.Lock()


def app_name_to_id(app_name: str, channel_name: Optional[str] = None):
    """Resolve (appName, channelName) → (appId, channelId); raises on miss."""
    app = storage.get_meta_data_apps().get_by_name(app_name)
    if app is None:
        raise ValueError(f"App name {app_name} is invalid.")
    if channel_name is None:
        return app.id, None
    for ch in storage.get_meta_data_channels().get_by_app_id(app.id):
        if ch.name == channel_name:
            return app.id, ch.id
    raise ValueError(
        f"Channel name {channel_name} is invalid for app {app_name}.")


def find(app_name: str,
         channel_name: Optional[str] = None,
         start_time: Optional[datetime] = None,
         until_time: Optional[datetime] = None,
         entity_type: Optional[str] = None,
         entity_id: Optional[str] = None,
         event_names: Optional[List[str]] = None,
         target_entity_type: Any = UNSET,
         target_entity_id: Any = UNSET,
         limit: Optional[int] = None,
         reversed: bool = False) -> List[Event]:
    """Training-time bulk read (PEventStore.find)."""
    app_id, channel_id = app_name_to_id(app_name, channel_name)
    return list(storage.get_p_events().find(
        app_id=app_id, channel_id=channel_id, start_time=start_time,
        until_time=until_time, entity_type=entity_type, entity_id=entity_id,
        event_names=event_names, target_entity_type=target_entity_type,
        target_entity_id=target_entity_id, limit=limit, reversed=reversed))


def find_columns(app_name: str,
                 channel_name: Optional[str] = None,
                 start_time: Optional[datetime] = None,
                 until_time: Optional[datetime] = None,
                 entity_type: Optional[str] = None,
                 event_names: Optional[List[str]] = None,
                 target_entity_type: Any = UNSET,
                 property_fields=()) -> Dict[str, list]:
    """Training-time bulk COLUMNAR read (PEventStore.find semantics with
    columnar output — the event-store→device ingest path; see
    LEvents.find_columns)."""
    app_id, channel_id = app_name_to_id(app_name, channel_name)
    return storage.get_p_events().find_columns(
        app_id=app_id, channel_id=channel_id, start_time=start_time,
        until_time=until_time, entity_type=entity_type,
        event_names=event_names, target_entity_type=target_entity_type,
        property_fields=property_fields)


def aggregate_properties(app_name: str, entity_type: str,
                         channel_name: Optional[str] = None,
                         start_time: Optional[datetime] = None,
                         until_time: Optional[datetime] = None,
                         required: Optional[List[str]] = None
                         ) -> Dict[str, PropertyMap]:
    """PEventStore.aggregateProperties."""
    app_id, channel_id = app_name_to_id(app_name, channel_name)
    return storage.get_p_events().aggregate_properties(
        app_id=app_id, channel_id=channel_id, entity_type=entity_type,
        start_time=start_time, until_time=until_time, required=required)


def find_by_entity(app_name: str, entity_type: str, entity_id: str,
                   channel_name: Optional[str] = None,
                   event_names: Optional[List[str]] = None,
                   target_entity_type: Any = UNSET,
                   target_entity_id: Any = UNSET,
                   start_time: Optional[datetime] = None,
                   until_time: Optional[datetime] = None,
                   limit: Optional[int] = None,
                   latest: bool = True,
                   timeout: float = 10.0) -> List[Event]:
    """Serving-time entity lookup with timeout
    (LEventStore.findByEntity, LEventStore.scala:76-106). Raises
    TimeoutError when the store does not answer within `timeout` seconds —
    templates catch this and degrade (the reference's 200 ms convention)."""
    app_id, channel_id = app_name_to_id(app_name, channel_name)

    def _q():
        return list(storage.get_l_events().find(
            app_id=app_id, channel_id=channel_id, entity_type=entity_type,
            entity_id=entity_id, event_names=event_names,
            target_entity_type=target_entity_type,
            target_entity_id=target_entity_id,
            start_time=start_time, until_time=until_time,
            limit=limit, reversed=latest))

    global _inflight
    with _inflight_lock:
        _inflight += 1
        inflight = _inflight
    if inflight > _POOL_WORKERS:
        logger.warning(
            "event_store lookup pool saturated: %d in-flight > %d workers "
            "(slow/locked store backing up serving-time lookups)",
            inflight, _POOL_WORKERS)

    def _done(_f):
        global _inflight
        with _inflight_lock:
            _inflight -= 1

    fut = _pool.submit(_q)
    fut.add_done_callback(_done)
    try:
        return fut.result(timeout=timeout)
    except concurrent.futures.TimeoutError as e:
        # if still queued, cancel so abandoned lookups don't consume a
        # worker; a query already running keeps its worker until the DB
        # answers (tracked by the _inflight gauge above)
        fut.cancel()
        raise TimeoutError(
            f"Event store lookup exceeded {timeout}s") from e
