"""Event data layer: Event model, DataMap, property aggregation, storage."""

from predictionio_amd.data.events import Event, DataMap, PropertyMap, validate_event
from predictionio_amd.data.bimap import BiMap

__all__ = ["Event", "DataMap", "PropertyMap", "validate_event", "BiMap"]
