"""SelfCleaningDataSource — event-window TTL + dedup + $set compression.

Parity with core/.../core/SelfCleaningDataSource.scala (324 LoC):
- EventWindow(duration, removeDuplicates, compressProperties)
- getCleanedEvents: drop events older than the window (:229 cleanPEvents),
  optionally remove exact duplicates (removeEvents :187), and compress a
  stream of `$set` events per entity into ONE event carrying the folded
  properties (compressPProperties :138-158)
- cleanPersistedEvents: rewrite the store with the cleaned set
  (cleanPersistedPEvents :160-175, wipe :176-186)

Mix into a DataSource:
    class MyDS(DataSource, SelfCleaningDataSource):
        app_name = "MyApp"
        event_window = EventWindow(duration=3600.0)
"""

from __future__ import annotations

import logging
from dataclasses import dataclass
from datetime import timedelta
from typing import List, Optional

from predictionio_amd.data import event_store, storage
from predictionio_amd.data.events import DataMap, Event, utcnow

log = logging.getLogger(__name__)


@dataclass
class EventWindow:
    """(SelfCleaningDataSource.scala EventWindow): duration in seconds
    (the reference parses strings like "3600 seconds")."""
    duration: Optional[float] = None
    remove_duplicates: bool = False
    compress_properties: bool = False


class SelfCleaningDataSource:
    app_name: str = ""
    channel_name: Optional[str] = None
    event_window: Optional[EventWindow] = None

    def _window_start(self):
        if self.event_window is None or self.event_window.duration is None:
            return None
        return utcnow() - timedelta(seconds=self.event_window.duration)

    def get_cleaned_events(self, events: List[Event]) -> List[Event]:
        """TTL filter + dedup + $set compression, in-memory."""
        w = self.event_window
        if w is None:
            return events
        start = self._window_start()
        if start is not None:
            events = [e for e in events if e.event_time >= start]
        if w.remove_duplicates:
            seen = set()
            out = []
            for e in events:
                k = (e.event, e.entity_type, e.entity_id,
                     e.target_entity_type, e.target_entity_id,
                     tuple(sorted(e.properties.to_dict().items())))
                if k not in seen:
                    seen.add(k)
                    out.append(e)
            events = out
        if w.compress_properties:
            events = self._compress_set_events(events)
        return events

    @staticmethod
    def _compress_set_events(events: List[Event]) -> List[Event]:
        """Fold consecutive $set events per entity into one
        (compressPProperties semantics: later values win)."""
        sets = {}
        order: List[Event] = []
        for e in sorted(events, key=lambda e: e.event_time):
            if e.event == "$set":
                k = (e.entity_type, e.entity_id)
                if k in sets:
                    merged = dict(sets[k].properties.to_dict())
                    merged.update(e.properties.to_dict())
                    sets[k] = Event(
                        event="$set", entity_type=e.entity_type,
                        entity_id=e.entity_id,
                        properties=DataMap(merged),
                        event_time=e.event_time)
                else:
                    sets[k] = e
            else:
                order.append(e)
        return list(sets.values()) + order

    def read_cleaned_events(self) -> List[Event]:
        """getCleanedPEvents path: read all + clean."""
        events = event_store.find(self.app_name,
                                  channel_name=self.channel_name)
        return self.get_cleaned_events(events)

    def clean_persisted_events(self) -> int:
        """Rewrite the event store with the cleaned set
        (cleanPersistedPEvents + wipe). Returns the number of events kept."""
        app_id, channel_id = event_store.app_name_to_id(
            self.app_name, self.channel_name)
        cleaned = self.read_cleaned_events()
        le = storage.get_l_events()
        le.remove(app_id, channel_id)
        le.init(app_id, channel_id)
        le.insert_batch(cleaned, app_id, channel_id)
        log.info("self-cleaning kept %d events for app %s",
                 len(cleaned), self.app_name)
        return len(cleaned)
