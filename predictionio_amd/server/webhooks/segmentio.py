"""Segment.io webhook connector.

Parity with data/.../webhooks/segmentio/SegmentIOConnector.scala:
- supports message types identify / track / alias / page / screen / group
  (SegmentIOConnector.scala:52-92)
- entity is always the user: user_id, falling back to anonymous_id
  (commonToJson, :170-184)
- properties = per-type payload, with the `context` object folded in when
  present (properties(), :153-165)
"""

from __future__ import annotations

from typing import Any, Dict

from predictionio_amd.server.webhooks import ConnectorException, JsonConnector


class SegmentIOConnector(JsonConnector):
    TYPES = {"identify", "track", "alias", "page", "screen", "group"}

    def to_event_json(self, data: Dict[str, Any]) -> Dict[str, Any]:
        if "version" not in data:
            raise ConnectorException(
                "Failed to get segment.io API version.")
        typ = data.get("type")
        if typ not in self.TYPES:
            raise ConnectorException(
                f"Cannot convert unknown type {typ} to event JSON.")

        user_id = data.get("user_id") or data.get("anonymous_id")
        if not user_id:
            raise ConnectorException(
                "there was no `userId` or `anonymousId` in the common fields.")

        if typ == "identify":
            props: Dict[str, Any] = {"traits": data.get("traits")}
        elif typ == "track":
            if "event" not in data:
                raise ConnectorException(
                    f"Cannot convert {data} to event JSON.")
            props = {"properties": data.get("properties"),
                     "event": data["event"]}
        elif typ == "alias":
            if "previous_id" not in data:
                raise ConnectorException(
                    f"Cannot convert {data} to event JSON.")
            props = {"previous_id": data["previous_id"]}
        elif typ in ("page", "screen"):
            props = {"name": data.get("name"),
                     "properties": data.get("properties")}
        else:  # group
            if "group_id" not in data:
                raise ConnectorException(
                    f"Cannot convert {data} to event JSON.")
            props = {"group_id": data["group_id"],
                     "traits": data.get("traits")}

        if data.get("context") is not None:
            props = {"context": data["context"], **props}

        out = {
            "event": typ,
            "entityType": "user",
            "entityId": user_id,
            "properties": props,
        }
        if data.get("timestamp"):
            out["eventTime"] = data["timestamp"]
        return out
