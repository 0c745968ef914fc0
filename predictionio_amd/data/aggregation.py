"""Property aggregation: fold $set/$unset/$delete streams into latest state.

Behavioral parity with the reference aggregators:
- local: data/.../storage/LEventAggregator.scala:42-60 (propAggregator)
- distributed: data/.../storage/PEventAggregator.scala

Semantics: events for one entity are folded in event-time order;
  $set   — union new properties over existing (new wins)
  $unset — remove the named keys
  $delete— drop the entity entirely (aggregation restarts if later $set)
Out-of-order events with older eventTime than the current state do not
overwrite newer values (reference folds in time order; we sort).
"""

from __future__ import annotations

from typing import Dict, Iterable, Optional, Tuple

from predictionio_amd.data.events import Event, PropertyMap


def aggregate_properties(events: Iterable[Event]) -> Dict[str, PropertyMap]:
    """Aggregate $set/$unset/$delete events into PropertyMap per entityId."""
    by_entity: Dict[str, list] = {}
    for e in events:
        if e.event in ("$set", "$unset", "$delete"):
            by_entity.setdefault(e.entity_id, []).append(e)
    out: Dict[str, PropertyMap] = {}
    for eid, evs in by_entity.items():
        pm = _fold(sorted(evs, key=lambda e: e.event_time))
        if pm is not None:
            out[eid] = pm
    return out


def _fold(evs) -> Optional[PropertyMap]:
    state: Optional[Tuple[dict, object, object]] = None  # (fields, first, last)
    for e in evs:
        if e.event == "$delete":
            state = None
        elif e.event == "$set":
            if state is None:
                state = (dict(e.properties.fields), e.event_time, e.event_time)
            else:
                fields, first, _ = state
                merged = dict(fields)
                merged.update(e.properties.fields)
                state = (merged, first, e.event_time)
        elif e.event == "$unset":
            if state is not None:
                fields, first, _ = state
                remaining = {k: v for k, v in fields.items()
                             if k not in e.properties.fields}
                state = (remaining, first, e.event_time)
    if state is None:
        return None
    fields, first, last = state
    return PropertyMap(fields, first, last)
