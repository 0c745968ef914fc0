"""pypio-compatible Python API.

Parity with python/pypio/pypio.py:31-111 (the reference's py4j bridge into
the JVM for PySpark/Jupyter users): init, find_events, save_model. Here
the framework IS Python, so these are thin wrappers over the native
storage/workflow layers — no bridge process. `find_events` returns the
event list (the reference returns a Spark DataFrame; callers here get
torch-friendly rows via `events_to_columns`).
"""

from __future__ import annotations

import pickle
from typing import Any, Dict, List, Optional

from predictionio_amd.data import event_store, storage
from predictionio_amd.data.events import Event, utcnow
from predictionio_amd.data.storage.base import EngineInstance, Model

_initialized = False


def init() -> None:
    """Initialize + verify the storage registry (pypio.init :31-44)."""
    global _initialized
    storage.verify_all_data_objects()
    _initialized = True


def find_events(app_name: str, channel_name: Optional[str] = None,
                **filters) -> List[Event]:
    """All events of an app (pypio.find_events :46-53)."""
    return event_store.find(app_name, channel_name=channel_name, **filters)


def events_to_columns(events: List[Event]) -> Dict[str, list]:
    """Columnar view (stand-in for the reference's DataFrame)."""
    return {
        "event": [e.event for e in events],
        "entityType": [e.entity_type for e in events],
        "entityId": [e.entity_id for e in events],
        "targetEntityType": [e.target_entity_type for e in events],
        "targetEntityId": [e.target_entity_id for e in events],
        "properties": [e.properties.to_dict() for e in events],
        "eventTime": [e.event_time for e in events],
    }


def save_model(model: Any, engine_factory: str = "pypio",
               engine_id: str = "pypio", engine_variant: str = "pypio",
               ) -> str:
    """Persist a model + COMPLETED EngineInstance record; returns the
    instance id (pypio.save_model :57-111 writes the same pair)."""
    instances = storage.get_meta_data_engine_instances()
    inst = EngineInstance(
        id="", status="INIT", start_time=utcnow(), end_time=utcnow(),
        engine_id=engine_id, engine_version="0",
        engine_variant=engine_variant, engine_factory=engine_factory)
    iid = instances.insert(inst)
    blob = pickle.dumps([model], protocol=pickle.HIGHEST_PROTOCOL)
    storage.get_model_data_models().insert(Model(iid, blob))
    inst.id = iid
    inst.status = "COMPLETED"
    inst.end_time = utcnow()
    instances.update(inst)
    return iid


def load_model(instance_id: str) -> Any:
    blob = storage.get_model_data_models().get(instance_id)
    if blob is None:
        raise KeyError(f"no model for instance {instance_id}")
    return pickle.loads(blob.models)[0]
