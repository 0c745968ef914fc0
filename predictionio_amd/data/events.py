"""Canonical event record, property bag and validation.

Behavioral parity with the reference event model:
- record fields: data/src/main/scala/.../storage/Event.scala:42-60
- validation rules: Event.scala:112-167 (EventValidation)
- DataMap typed accessors: data/.../storage/DataMap.scala:45-245
- PropertyMap first/lastUpdated: data/.../storage/PropertyMap.scala:30-99
- wire schema: data/.../storage/EventJson4sSupport.scala:44-240

Implementation is new (Python dataclasses + stdlib datetime); no JVM/json4s.
"""

from __future__ import annotations

import uuid
from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Any, Dict, Iterator, List, Optional

RESERVED_PREFIXES = ("$", "pio_")
SPECIAL_EVENTS = frozenset({"$set", "$unset", "$delete"})
BUILTIN_ENTITY_TYPES = frozenset({"pio_pr"})
BUILTIN_PROPERTIES: frozenset = frozenset()


def utcnow() -> datetime:
    return datetime.now(timezone.utc)


def parse_time(value: Any) -> datetime:
    """Parse an ISO8601 timestamp (reference: DateTimeJson4sSupport, UTC default)."""
    if isinstance(value, datetime):
        return value if value.tzinfo else value.replace(tzinfo=timezone.utc)
    if isinstance(value, (int, float)):
        return datetime.fromtimestamp(value / 1000.0, tz=timezone.utc)
    s = str(value)
    if s.endswith("Z"):
        s = s[:-1] + "+00:00"
    dt = datetime.fromisoformat(s)
    return dt if dt.tzinfo else dt.replace(tzinfo=timezone.utc)


def format_time(dt: datetime) -> str:
    """ISO8601 with millisecond precision, matching the reference wire format."""
    if dt.tzinfo is None:
        dt = dt.replace(tzinfo=timezone.utc)
    return dt.isoformat(timespec="milliseconds")


class DataMap:
    """JSON-valued property bag with typed access (DataMap.scala:45-245)."""

    __slots__ = ("fields",)

    def __init__(self, fields: Optional[Dict[str, Any]] = None):
        self.fields: Dict[str, Any] = dict(fields or {})

    def require(self, name: str) -> None:
        if name not in self.fields:
            raise KeyError(f"The field {name} is required.")

    def contains(self, name: str) -> bool:
        return name in self.fields

    def __contains__(self, name: str) -> bool:
        return name in self.fields

    def get(self, name: str, typ: Optional[type] = None) -> Any:
        """Mandatory typed accessor — raises if absent or null."""
        self.require(name)
        v = self.fields[name]
        if v is None:
            raise ValueError(f"The field {name} cannot be null.")
        return _coerce(v, typ) if typ else v

    def get_opt(self, name: str, typ: Optional[type] = None) -> Any:
        if name not in self.fields or self.fields[name] is None:
            return None
        v = self.fields[name]
        return _coerce(v, typ) if typ else v

    def get_or_else(self, name: str, default: Any) -> Any:
        v = self.get_opt(name)
        return default if v is None else v

    def keySet(self) -> set:
        return set(self.fields.keys())

    @property
    def is_empty(self) -> bool:
        return not self.fields

    def __iter__(self) -> Iterator[str]:
        return iter(self.fields)

    def __eq__(self, other) -> bool:
        return isinstance(other, DataMap) and self.fields == other.fields

    def __repr__(self) -> str:
        return f"DataMap({self.fields!r})"

    def union(self, other: "DataMap") -> "DataMap":
        merged = dict(self.fields)
        merged.update(other.fields)
        return DataMap(merged)

    def minus(self, keys) -> "DataMap":
        return DataMap({k: v for k, v in self.fields.items() if k not in set(keys)})

    def to_dict(self) -> Dict[str, Any]:
        return dict(self.fields)


def _coerce(v: Any, typ: type) -> Any:
    if typ is float and isinstance(v, (int, float)):
        return float(v)
    if typ is int and isinstance(v, (int, float)) and float(v).is_integer():
        return int(v)
    if typ is str:
        return v if isinstance(v, str) else str(v)
    if typ is bool:
        if isinstance(v, bool):
            return v
        raise TypeError(f"field is not a bool: {v!r}")
    if typ is list:
        if isinstance(v, list):
            return v
        raise TypeError(f"field is not a list: {v!r}")
    if not isinstance(v, typ):
        raise TypeError(f"field has type {type(v).__name__}, expected {typ.__name__}")
    return v


class PropertyMap(DataMap):
    """DataMap plus aggregation bookkeeping (PropertyMap.scala:30-99)."""

    __slots__ = ("first_updated", "last_updated")

    def __init__(self, fields: Optional[Dict[str, Any]],
                 first_updated: datetime, last_updated: datetime):
        super().__init__(fields)
        self.first_updated = first_updated
        self.last_updated = last_updated

    def __repr__(self) -> str:
        return (f"PropertyMap({self.fields!r}, first={self.first_updated}, "
                f"last={self.last_updated})")


@dataclass
class Event:
    """Canonical event record (Event.scala:42-60)."""

    event: str
    entity_type: str
    entity_id: str
    target_entity_type: Optional[str] = None
    target_entity_id: Optional[str] = None
    properties: DataMap = field(default_factory=DataMap)
    event_time: datetime = field(default_factory=utcnow)
    tags: List[str] = field(default_factory=list)
    pr_id: Optional[str] = None
    creation_time: datetime = field(default_factory=utcnow)
    event_id: Optional[str] = None

    def to_json(self, with_id: bool = True) -> Dict[str, Any]:
        """Wire schema of the reference (EventJson4sSupport.writeJson)."""
        d: Dict[str, Any] = {}
        if with_id and self.event_id is not None:
            d["eventId"] = self.event_id
        d.update({
            "event": self.event,
            "entityType": self.entity_type,
            "entityId": self.entity_id,
        })
        if self.target_entity_type is not None:
            d["targetEntityType"] = self.target_entity_type
        if self.target_entity_id is not None:
            d["targetEntityId"] = self.target_entity_id
        d["properties"] = self.properties.to_dict()
        d["eventTime"] = format_time(self.event_time)
        if self.tags:
            d["tags"] = list(self.tags)
        if self.pr_id is not None:
            d["prId"] = self.pr_id
        d["creationTime"] = format_time(self.creation_time)
        return d

    @staticmethod
    def from_json(d: Dict[str, Any]) -> "Event":
        """Wire schema reader (EventJson4sSupport.readJson semantics):
        `event`, `entityType`, `entityId` mandatory; times default to now."""
        for k in ("event", "entityType", "entityId"):
            if k not in d:
                raise ValueError(f"field {k} is required")
            if not isinstance(d[k], str):
                raise ValueError(f"field {k} must be a string")
        now = utcnow()
        props = d.get("properties", {})
        if props is None:
            props = {}
        if not isinstance(props, dict):
            raise ValueError("properties must be a JSON object")
        tags = d.get("tags", [])
        if tags is None:
            tags = []
        return Event(
            event=d["event"],
            entity_type=d["entityType"],
            entity_id=d["entityId"],
            target_entity_type=d.get("targetEntityType"),
            target_entity_id=d.get("targetEntityId"),
            properties=DataMap(props),
            event_time=parse_time(d["eventTime"]) if d.get("eventTime") else now,
            tags=list(tags),
            pr_id=d.get("prId"),
            creation_time=parse_time(d["creationTime"]) if d.get("creationTime") else now,
            event_id=d.get("eventId"),
        )


def is_reserved_prefix(name: str) -> bool:
    return name.startswith("$") or name.startswith("pio_")


def is_special_event(name: str) -> bool:
    return name in SPECIAL_EVENTS


def validate_event(e: Event) -> None:
    """Validation rules of the reference (Event.scala:112-167). Raises ValueError."""
    def require(cond: bool, msg: str) -> None:
        if not cond:
            raise ValueError(msg)

    require(bool(e.event), "event must not be empty.")
    require(bool(e.entity_type), "entityType must not be empty string.")
    require(bool(e.entity_id), "entityId must not be empty string.")
    require(e.target_entity_type is None or bool(e.target_entity_type),
            "targetEntityType must not be empty string")
    require(e.target_entity_id is None or bool(e.target_entity_id),
            "targetEntityId must not be empty string.")
    require(not ((e.target_entity_type is not None) and (e.target_entity_id is None)),
            "targetEntityType and targetEntityId must be specified together.")
    require(not ((e.target_entity_type is None) and (e.target_entity_id is not None)),
            "targetEntityType and targetEntityId must be specified together.")
    require(not (e.event == "$unset" and e.properties.is_empty),
            "properties cannot be empty for $unset event")
    require(not is_reserved_prefix(e.event) or is_special_event(e.event),
            f"{e.event} is not a supported reserved event name.")
    require(not is_special_event(e.event) or
            (e.target_entity_type is None and e.target_entity_id is None),
            f"Reserved event {e.event} cannot have targetEntity")
    require(not is_reserved_prefix(e.entity_type) or
            e.entity_type in BUILTIN_ENTITY_TYPES,
            f"The entityType {e.entity_type} is not allowed. "
            f"'pio_' is a reserved name prefix.")
    if e.target_entity_type is not None:
        require(not is_reserved_prefix(e.target_entity_type) or
                e.target_entity_type in BUILTIN_ENTITY_TYPES,
                f"The targetEntityType {e.target_entity_type} is not allowed. "
                f"'pio_' is a reserved name prefix.")
    for k in e.properties.keySet():
        require(not is_reserved_prefix(k) or k in BUILTIN_PROPERTIES,
                f"The property {k} is not allowed. 'pio_' is a reserved name prefix.")


def new_event_id() -> str:
    return uuid.uuid4().hex
