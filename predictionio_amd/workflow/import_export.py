"""Event import/export as JSON lines.

Parity with tools/.../imprt/FileToEvents.scala:40-112 (JSON-lines file →
event store, with validation) and tools/.../export/EventsToFile.scala
(event store → JSON-lines file). The reference runs these as Spark jobs;
here they stream through the storage DAO directly.
"""

from __future__ import annotations

import json
from typing import Optional

from predictionio_amd.data import storage
from predictionio_amd.data.events import Event, validate_event


def _channel_id(app_id: int, channel: Optional[str]) -> Optional[int]:
    if channel is None:
        return None
    chs = [c for c in storage.get_meta_data_channels().get_by_app_id(app_id)
           if c.name == channel]
    if not chs:
        raise ValueError(f"Channel {channel} not found for app {app_id}")
    return chs[0].id


def import_events(app_id: int, input_path: str,
                  channel: Optional[str] = None,
                  batch_size: int = 1000) -> int:
    """JSON-lines → event store; every event validated before insert."""
    le = storage.get_l_events()
    cid = _channel_id(app_id, channel)
    le.init(app_id, cid)
    n = 0
    batch = []
    with open(input_path) as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            e = Event.from_json(json.loads(line))
            validate_event(e)
            batch.append(e)
            if len(batch) >= batch_size:
                le.insert_batch(batch, app_id, cid)
                n += len(batch)
                batch = []
    if batch:
        le.insert_batch(batch, app_id, cid)
        n += len(batch)
    return n


def export_events(app_id: int, output_path: str,
                  channel: Optional[str] = None) -> int:
    """Event store → JSON-lines (EventsToFile json format)."""
    le = storage.get_l_events()
    cid = _channel_id(app_id, channel)
    n = 0
    with open(output_path, "w") as out:
        for e in le.find(app_id, cid):
            out.write(json.dumps(e.to_json()) + "\n")
            n += 1
    return n
