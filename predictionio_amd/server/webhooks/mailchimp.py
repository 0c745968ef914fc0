"""MailChimp webhook connector (form-encoded payloads).

Parity with data/.../webhooks/mailchimp/MailChimpConnector.scala:
types subscribe / unsubscribe / profile / upemail / cleaned / campaign
(:37-56) with the same event/entity/property mapping per type and
`fired_at` ("yyyy-MM-dd HH:mm:ss", UTC) converted to ISO8601 (:60-63).
"""

from __future__ import annotations

from datetime import datetime, timezone
from typing import Any, Dict, Mapping

from predictionio_amd.server.webhooks import ConnectorException, FormConnector


def _parse_time(s: str) -> str:
    dt = datetime.strptime(s, "%Y-%m-%d %H:%M:%S").replace(
        tzinfo=timezone.utc)
    return dt.isoformat()


class MailChimpConnector(FormConnector):
    def to_event_json(self, data: Mapping[str, str]) -> Dict[str, Any]:
        typ = data.get("type")
        handlers = {
            "subscribe": self._subscribe,
            "unsubscribe": self._unsubscribe,
            "profile": self._profile,
            "upemail": self._upemail,
            "cleaned": self._cleaned,
            "campaign": self._campaign,
        }
        if typ is None:
            raise ConnectorException(
                "The field 'type' is required for MailChimp data.")
        if typ not in handlers:
            raise ConnectorException(
                f"Cannot convert unknown MailChimp data type {typ} "
                "to event JSON")
        try:
            return handlers[typ](data)
        except KeyError as e:
            raise ConnectorException(
                f"Cannot convert MailChimp data to event JSON: "
                f"missing field {e}") from e

    @staticmethod
    def _merges(data: Mapping[str, str]) -> Dict[str, Any]:
        return {
            "EMAIL": data["data[merges][EMAIL]"],
            "FNAME": data["data[merges][FNAME]"],
            "LNAME": data["data[merges][LNAME]"],
            "INTERESTS": data.get("data[merges][INTERESTS]"),
        }

    def _subscribe(self, d: Mapping[str, str]) -> Dict[str, Any]:
        return {
            "event": "subscribe",
            "entityType": "user",
            "entityId": d["data[id]"],
            "targetEntityType": "list",
            "targetEntityId": d["data[list_id]"],
            "eventTime": _parse_time(d["fired_at"]),
            "properties": {
                "email": d["data[email]"],
                "email_type": d["data[email_type]"],
                "merges": self._merges(d),
                "ip_opt": d["data[ip_opt]"],
                "ip_signup": d["data[ip_signup]"],
            },
        }

    def _unsubscribe(self, d: Mapping[str, str]) -> Dict[str, Any]:
        return {
            "event": "unsubscribe",
            "entityType": "user",
            "entityId": d["data[id]"],
            "targetEntityType": "list",
            "targetEntityId": d["data[list_id]"],
            "eventTime": _parse_time(d["fired_at"]),
            "properties": {
                "action": d["data[action]"],
                "reason": d["data[reason]"],
                "email": d["data[email]"],
                "email_type": d["data[email_type]"],
                "merges": self._merges(d),
                "ip_opt": d["data[ip_opt]"],
                "campaign_id": d["data[campaign_id]"],
            },
        }

    def _profile(self, d: Mapping[str, str]) -> Dict[str, Any]:
        return {
            "event": "profile",
            "entityType": "user",
            "entityId": d["data[id]"],
            "targetEntityType": "list",
            "targetEntityId": d["data[list_id]"],
            "eventTime": _parse_time(d["fired_at"]),
            "properties": {
                "email": d["data[email]"],
                "email_type": d["data[email_type]"],
                "merges": self._merges(d),
                "ip_opt": d["data[ip_opt]"],
            },
        }

    def _upemail(self, d: Mapping[str, str]) -> Dict[str, Any]:
        return {
            "event": "upemail",
            "entityType": "user",
            "entityId": d["data[new_id]"],
            "targetEntityType": "list",
            "targetEntityId": d["data[list_id]"],
            "eventTime": _parse_time(d["fired_at"]),
            "properties": {
                "new_email": d["data[new_email]"],
                "old_email": d["data[old_email]"],
            },
        }

    def _cleaned(self, d: Mapping[str, str]) -> Dict[str, Any]:
        return {
            "event": "cleaned",
            "entityType": "list",
            "entityId": d["data[list_id]"],
            "eventTime": _parse_time(d["fired_at"]),
            "properties": {
                "campaignId": d["data[campaign_id]"],
                "reason": d["data[reason]"],
                "email": d["data[email]"],
            },
        }

    def _campaign(self, d: Mapping[str, str]) -> Dict[str, Any]:
        return {
            "event": "campaign",
            "entityType": "campaign",
            "entityId": d["data[id]"],
            "targetEntityType": "list",
            "targetEntityId": d["data[list_id]"],
            "eventTime": _parse_time(d["fired_at"]),
            "properties": {
                "subject": d["data[subject]"],
                "status": d["data[status]"],
                "reason": d["data[reason]"],
            },
        }
