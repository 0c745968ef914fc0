"""Example webhook connectors (development references).

Parity with data/.../webhooks/examplejson/ExampleJsonConnector.scala and
exampleform/ExampleFormConnector.scala: payload types `userAction` and
`userActionItem` mapped onto user / user→item events.
"""

from __future__ import annotations

from typing import Any, Dict, Mapping

from predictionio_amd.server.webhooks import (
    ConnectorException, FormConnector, JsonConnector,
)


class ExampleJsonConnector(JsonConnector):
    def to_event_json(self, data: Dict[str, Any]) -> Dict[str, Any]:
        typ = data.get("type")
        if typ is None:
            raise ConnectorException("Cannot extract Common field.")
        try:
            if typ == "userAction":
                return {
                    "event": data["event"],
                    "entityType": "user",
                    "entityId": data["userId"],
                    "eventTime": data["timestamp"],
                    "properties": {
                        "context": data.get("context"),
                        "anotherProperty1": data["anotherProperty1"],
                        "anotherProperty2": data.get("anotherProperty2"),
                    },
                }
            if typ == "userActionItem":
                return {
                    "event": data["event"],
                    "entityType": "user",
                    "entityId": data["userId"],
                    "targetEntityType": "item",
                    "targetEntityId": data["itemId"],
                    "eventTime": data["timestamp"],
                    "properties": {
                        "context": data.get("context"),
                        "anotherPropertyA": data.get("anotherPropertyA"),
                        "anotherPropertyB": data.get("anotherPropertyB"),
                    },
                }
        except KeyError as e:
            raise ConnectorException(
                f"Cannot convert {data} to eventJson. Missing {e}.") from e
        raise ConnectorException(
            f"Cannot convert unknown type '{typ}' to Event JSON.")


class ExampleFormConnector(FormConnector):
    def to_event_json(self, data: Mapping[str, str]) -> Dict[str, Any]:
        typ = data.get("type")
        if typ is None:
            raise ConnectorException("The field 'type' is required.")
        try:
            if typ == "userAction":
                context = None
                if any(k.startswith("context[") for k in data):
                    context = {
                        "ip": data.get("context[ip]"),
                        "prop1": (float(data["context[prop1]"])
                                  if "context[prop1]" in data else None),
                        "prop2": data.get("context[prop2]"),
                    }
                return {
                    "event": data["event"],
                    "entityType": "user",
                    "entityId": data["userId"],
                    "eventTime": data["timestamp"],
                    "properties": {
                        "context": context,
                        "anotherProperty1": int(data["anotherProperty1"]),
                        "anotherProperty2": data.get("anotherProperty2"),
                    },
                }
            if typ == "userActionItem":
                return {
                    "event": data["event"],
                    "entityType": "user",
                    "entityId": data["userId"],
                    "targetEntityType": "item",
                    "targetEntityId": data["itemId"],
                    "eventTime": data["timestamp"],
                    "properties": {
                        "context": {
                            "ip": data["context[ip]"],
                            "prop1": float(data["context[prop1]"]),
                            "prop2": data["context[prop2]"],
                        },
                        "anotherPropertyA": (
                            float(data["anotherPropertyA"])
                            if "anotherPropertyA" in data else None),
                        "anotherPropertyB": (
                            data["anotherPropertyB"] == "true"
                            if "anotherPropertyB" in data else None),
                    },
                }
        except (KeyError, ValueError) as e:
            raise ConnectorException(
                f"Cannot convert {dict(data)} to event JSON. {e}") from e
        raise ConnectorException(
            f"Cannot convert unknown type {typ} to event JSON")
