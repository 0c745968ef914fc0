"""Webhooks connector framework.

Parity with the reference webhooks layer (data/.../webhooks/):
- JsonConnector.toEventJson(JObject) (webhooks/JsonConnector.scala:27-35)
- FormConnector.toEventJson(Map[String,String]) (webhooks/FormConnector.scala)
- dispatch table WebhooksConnectors.json/form (WebhooksConnectors.scala:27-37)
- POST/GET handling incl. per-connector paths (api/Webhooks.scala:32-138)

A connector translates a third-party payload into the canonical event JSON
(the dict accepted by Event.from_json); the event server then validates and
stores it like any other event.
"""

from __future__ import annotations

from typing import Any, Dict, Mapping


class ConnectorException(Exception):
    """Raised when a payload cannot be mapped to an event
    (webhooks/ConnectorException.scala)."""


class JsonConnector:
    """JSON-payload connector (JsonConnector.scala:27-35)."""

    def to_event_json(self, data: Dict[str, Any]) -> Dict[str, Any]:
        raise NotImplementedError


class FormConnector:
    """Form-encoded-payload connector (FormConnector.scala:27-35)."""

    def to_event_json(self, data: Mapping[str, str]) -> Dict[str, Any]:
        raise NotImplementedError


def json_connectors() -> Dict[str, JsonConnector]:
    """web-path → connector (WebhooksConnectors.json, :27-31)."""
    from predictionio_amd.server.webhooks.segmentio import SegmentIOConnector
    from predictionio_amd.server.webhooks.examples import ExampleJsonConnector
    return {
        "segmentio": SegmentIOConnector(),
        "examplejson": ExampleJsonConnector(),
    }


def form_connectors() -> Dict[str, FormConnector]:
    """web-path → connector (WebhooksConnectors.form, :33-37)."""
    from predictionio_amd.server.webhooks.mailchimp import MailChimpConnector
    from predictionio_amd.server.webhooks.examples import ExampleFormConnector
    return {
        "mailchimp": MailChimpConnector(),
        "exampleform": ExampleFormConnector(),
    }
