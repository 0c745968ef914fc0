"""Storage DAO contracts + metadata records.

Parity with the reference storage abstraction layer:
- LEvents trait: data/.../storage/LEvents.scala:40-238 (insert/get/delete/
  find with 9 filter dimensions/aggregateProperties)
- metadata DAOs: Apps.scala:29-61, AccessKeys.scala:28-77, Channels.scala:28-82,
  EngineInstances.scala:46-98, EvaluationInstances.scala:33-138, Models.scala:32-51

The reference's async (Future-based) local API collapses to a synchronous
Python API: the new framework's event server offloads to a thread pool at the
HTTP layer instead of per-DAO futures.
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from datetime import datetime
from typing import Any, Dict, Iterable, List, Optional, Sequence

from predictionio_amd.data.events import Event, PropertyMap, utcnow


class _Unset:
    """Sentinel distinguishing "no filter" from "must be None" — the
    reference models this as Option[Option[String]] (LEvents.scala:188-200)."""

    def __repr__(self):
        return "UNSET"


UNSET = _Unset()


# ---------------------------------------------------------------- metadata

@dataclass
class App:
    id: int
    name: str
    description: Optional[str] = None


@dataclass
class AccessKey:
    key: str
    appid: int
    events: List[str] = field(default_factory=list)  # empty = all events allowed


@dataclass
class Channel:
    id: int
    name: str
    appid: int

    NAME_RE = re.compile(r"^[a-zA-Z0-9-]{1,16}$")

    @staticmethod
    def is_valid_name(s: str) -> bool:
        """Channel name constraint (Channels.scala:49-56)."""
        return bool(Channel.NAME_RE.match(s or ""))


@dataclass
class EngineInstance:
    """Record of one training run (EngineInstances.scala:46-98)."""
    id: str
    status: str  # INIT | TRAINING | COMPLETED | FAILED
    start_time: datetime
    end_time: datetime
    engine_id: str
    engine_version: str
    engine_variant: str
    engine_factory: str
    batch: str = ""
    env: Dict[str, str] = field(default_factory=dict)
    runtime_conf: Dict[str, str] = field(default_factory=dict)
    data_source_params: str = ""
    preparator_params: str = ""
    algorithms_params: str = ""
    serving_params: str = ""


@dataclass
class EvaluationInstance:
    """Record of one evaluation run (EvaluationInstances.scala:33-138)."""
    id: str
    status: str
    start_time: datetime
    end_time: datetime
    evaluation_class: str = ""
    engine_params_generator_class: str = ""
    batch: str = ""
    env: Dict[str, str] = field(default_factory=dict)
    evaluator_results: str = ""
    evaluator_results_html: str = ""
    evaluator_results_json: str = ""


@dataclass
class Model:
    """Serialized model blob keyed by engine instance id (Models.scala:32-51)."""
    id: str
    models: bytes


# ---------------------------------------------------------------- DAO traits

class Apps:
    def insert(self, app: App) -> Optional[int]:
        raise NotImplementedError

    def get(self, app_id: int) -> Optional[App]:
        raise NotImplementedError

    def get_by_name(self, name: str) -> Optional[App]:
        raise NotImplementedError

    def get_all(self) -> List[App]:
        raise NotImplementedError

    def update(self, app: App) -> bool:
        raise NotImplementedError

    def delete(self, app_id: int) -> bool:
        raise NotImplementedError


class AccessKeys:
    def insert(self, k: AccessKey) -> Optional[str]:
        raise NotImplementedError

    def get(self, key: str) -> Optional[AccessKey]:
        raise NotImplementedError

    def get_all(self) -> List[AccessKey]:
        raise NotImplementedError

    def get_by_app_id(self, app_id: int) -> List[AccessKey]:
        raise NotImplementedError

    def update(self, k: AccessKey) -> bool:
        raise NotImplementedError

    def delete(self, key: str) -> bool:
        raise NotImplementedError


class Channels:
    def insert(self, c: Channel) -> Optional[int]:
        raise NotImplementedError

    def get(self, channel_id: int) -> Optional[Channel]:
        raise NotImplementedError

    def get_by_app_id(self, app_id: int) -> List[Channel]:
        raise NotImplementedError

    def delete(self, channel_id: int) -> bool:
        raise NotImplementedError


class EngineInstances:
    def insert(self, i: EngineInstance) -> str:
        raise NotImplementedError

    def get(self, iid: str) -> Optional[EngineInstance]:
        raise NotImplementedError

    def get_all(self) -> List[EngineInstance]:
        raise NotImplementedError

    def get_latest_completed(self, engine_id: str, engine_version: str,
                             engine_variant: str) -> Optional[EngineInstance]:
        raise NotImplementedError

    def get_latest_completed_by_factory(
            self, engine_factory: str,
            engine_variant: Optional[str] = None
    ) -> Optional["EngineInstance"]:
        """Deploy-time resolution (commands/Engine.deploy :208-245 /
        EngineInstances.getLatestCompleted EngineInstances.scala:69) —
        in the DAO so backends can push the filter into the store
        instead of the server scanning get_all(). Default: scan."""
        cands = [i for i in self.get_all()
                 if i.status == "COMPLETED"
                 and i.engine_factory == engine_factory
                 and (engine_variant is None
                      or i.engine_variant == engine_variant)]
        cands.sort(key=lambda i: i.start_time)
        return cands[-1] if cands else None

    def get_completed(self, engine_id: str, engine_version: str,
                      engine_variant: str) -> List[EngineInstance]:
        raise NotImplementedError

    def update(self, i: EngineInstance) -> bool:
        raise NotImplementedError

    def delete(self, iid: str) -> bool:
        raise NotImplementedError


class EvaluationInstances:
    def insert(self, i: EvaluationInstance) -> str:
        raise NotImplementedError

    def get(self, iid: str) -> Optional[EvaluationInstance]:
        raise NotImplementedError

    def get_all(self) -> List[EvaluationInstance]:
        raise NotImplementedError

    def get_completed(self) -> List[EvaluationInstance]:
        raise NotImplementedError

    def update(self, i: EvaluationInstance) -> bool:
        raise NotImplementedError

    def delete(self, iid: str) -> bool:
        raise NotImplementedError


class Models:
    def insert(self, m: Model) -> None:
        raise NotImplementedError

    def get(self, mid: str) -> Optional[Model]:
        raise NotImplementedError

    def delete(self, mid: str) -> bool:
        raise NotImplementedError


class LEvents:
    """Event CRUD + find + aggregate (LEvents.scala:40-238).

    All methods take app_id and optional channel_id; channel None = default.
    """

    def init(self, app_id: int, channel_id: Optional[int] = None) -> bool:
        raise NotImplementedError

    def remove(self, app_id: int, channel_id: Optional[int] = None) -> bool:
        raise NotImplementedError

    def close(self) -> None:
        pass

    def insert(self, event: Event, app_id: int,
               channel_id: Optional[int] = None) -> str:
        raise NotImplementedError

    def insert_batch(self, events: Sequence[Event], app_id: int,
                     channel_id: Optional[int] = None) -> List[str]:
        return [self.insert(e, app_id, channel_id) for e in events]

    def get(self, event_id: str, app_id: int,
            channel_id: Optional[int] = None) -> Optional[Event]:
        raise NotImplementedError

    def delete(self, event_id: str, app_id: int,
               channel_id: Optional[int] = None) -> bool:
        raise NotImplementedError

    def find(self, app_id: int, channel_id: Optional[int] = None,
             start_time: Optional[datetime] = None,
             until_time: Optional[datetime] = None,
             entity_type: Optional[str] = None,
             entity_id: Optional[str] = None,
             event_names: Optional[List[str]] = None,
             target_entity_type: Any = UNSET,
             target_entity_id: Any = UNSET,
             limit: Optional[int] = None,
             reversed: bool = False) -> Iterable[Event]:
        """9-dimension filter (LEvents.futureFind, LEvents.scala:188-200).

        target_entity_type/-id: UNSET = any; None = must be absent;
        a string = must equal. limit None = all, -1 = all. reversed sorts
        by eventTime descending."""
        raise NotImplementedError

    def find_columns(self, app_id: int, channel_id: Optional[int] = None,
                     start_time: Optional[datetime] = None,
                     until_time: Optional[datetime] = None,
                     entity_type: Optional[str] = None,
                     event_names: Optional[List[str]] = None,
                     target_entity_type: Any = UNSET,
                     property_fields: Sequence[str] = ()
                     ) -> Dict[str, list]:
        """Bulk columnar read for training ingest — PEvents.find
        semantics (PEvents.scala:80-89) returning parallel COLUMNS
        instead of per-event objects, so 10^7-10^9-event reads skip
        Python object construction entirely. Returns
        {'event': [str], 'entity_id': [str], 'target_entity_id':
        [str|None], 'event_time_ms': [int]} plus one column per
        requested property field (value or None). Rows are
        eventTime-ascending (the dedup contracts — latest-wins — rely
        on it). Backends override with store-side columnar scans; this
        default walks find()."""
        out: Dict[str, list] = {"event": [], "entity_id": [],
                                "target_entity_id": [], "event_time_ms": []}
        for f in property_fields:
            out[f] = []
        for e in self.find(app_id=app_id, channel_id=channel_id,
                           start_time=start_time, until_time=until_time,
                           entity_type=entity_type, event_names=event_names,
                           target_entity_type=target_entity_type):
            out["event"].append(e.event)
            out["entity_id"].append(e.entity_id)
            out["target_entity_id"].append(e.target_entity_id)
            out["event_time_ms"].append(
                int(e.event_time.timestamp() * 1000))
            for f in property_fields:
                out[f].append(e.properties.get_opt(f))
        return out

    def aggregate_properties(self, app_id: int, entity_type: str,
                             channel_id: Optional[int] = None,
                             start_time: Optional[datetime] = None,
                             until_time: Optional[datetime] = None,
                             required: Optional[List[str]] = None
                             ) -> Dict[str, PropertyMap]:
        """Fold $set/$unset/$delete into PropertyMap per entity
        (LEvents.futureAggregateProperties, LEvents.scala:215-238)."""
        from predictionio_amd.data.aggregation import aggregate_properties
        events = self.find(
            app_id=app_id, channel_id=channel_id,
            start_time=start_time, until_time=until_time,
            entity_type=entity_type,
            event_names=["$set", "$unset", "$delete"])
        result = aggregate_properties(events)
        if required:
            result = {k: v for k, v in result.items()
                      if all(r in v.fields for r in required)}
        return result
