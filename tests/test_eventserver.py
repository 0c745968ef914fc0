"""Event Server REST tests (reference: EventServiceSpec.scala,
SegmentIOAuthSpec.scala, webhook connector specs — akka-http-testkit
there, starlette TestClient here)."""

import pytest
from fastapi.testclient import TestClient

from predictionio_amd.data import storage
from predictionio_amd.data.storage.base import AccessKey, App, Channel


@pytest.fixture()
def client(tmp_path, monkeypatch):
    monkeypatch.setenv("PIO_FS_BASEDIR", str(tmp_path))
    monkeypatch.setenv("PIO_STORAGE_SQLITE_PATH", str(tmp_path / "pio.db"))
    storage.reset()
    apps = storage.get_meta_data_apps()
    app_id = apps.insert(App(id=0, name="testapp"))
    storage.get_meta_data_access_keys().insert(
        AccessKey(key="SECRET", appid=app_id, events=[]))
    storage.get_meta_data_access_keys().insert(
        AccessKey(key="LIMITED", appid=app_id, events=["buy"]))
    storage.get_meta_data_channels().insert(
        Channel(id=0, name="chan1", appid=app_id))
    storage.get_l_events().init(app_id)
    ch = storage.get_meta_data_channels().get_by_app_id(app_id)[0]
    storage.get_l_events().init(app_id, ch.id)

    from predictionio_amd.server.eventserver import create_app
    c = TestClient(create_app(stats_on=True))
    yield c
    storage.reset()


EV = {"event": "rate", "entityType": "user", "entityId": "u1",
      "targetEntityType": "item", "targetEntityId": "i3",
      "properties": {"rating": 4.5},
      "eventTime": "2026-01-02T03:04:05.000Z"}


class TestAuth:
    def test_root_alive(self, client):
        r = client.get("/")
        assert r.status_code == 200 and r.json() == {"status": "alive"}

    def test_missing_key(self, client):
        assert client.post("/events.json", json=EV).status_code == 401

    def test_bad_key(self, client):
        r = client.post("/events.json?accessKey=WRONG", json=EV)
        assert r.status_code == 401

    def test_basic_auth_header(self, client):
        import base64
        h = {"Authorization": "Basic " + base64.b64encode(b"SECRET:").decode()}
        r = client.post("/events.json", json=EV, headers=h)
        assert r.status_code == 201

    def test_bad_channel(self, client):
        r = client.post("/events.json?accessKey=SECRET&channel=nope",
                        json=EV)
        assert r.status_code == 401


class TestEventsCRUD:
    def test_post_get_delete(self, client):
        r = client.post("/events.json?accessKey=SECRET", json=EV)
        assert r.status_code == 201
        eid = r.json()["eventId"]
        r = client.get(f"/events/{eid}.json?accessKey=SECRET")
        assert r.status_code == 200
        got = r.json()
        assert got["event"] == "rate" and got["entityId"] == "u1"
        assert got["properties"] == {"rating": 4.5}
        r = client.delete(f"/events/{eid}.json?accessKey=SECRET")
        assert r.status_code == 200 and r.json() == {"message": "Found"}
        r = client.get(f"/events/{eid}.json?accessKey=SECRET")
        assert r.status_code == 404

    def test_restricted_key(self, client):
        r = client.post("/events.json?accessKey=LIMITED", json=EV)
        assert r.status_code == 403
        assert "not allowed" in r.json()["message"]
        buy = dict(EV, event="buy")
        assert client.post("/events.json?accessKey=LIMITED",
                           json=buy).status_code == 201

    def test_malformed(self, client):
        r = client.post("/events.json?accessKey=SECRET",
                        json={"event": "rate"})  # missing entity fields
        assert r.status_code == 400

    def test_reserved_event_validation(self, client):
        bad = dict(EV, event="$set", properties={})
        # $set with targetEntity is invalid (EventValidation.scala)
        assert client.post("/events.json?accessKey=SECRET",
                           json=bad).status_code == 400

    def test_channel_isolation(self, client):
        r = client.post("/events.json?accessKey=SECRET&channel=chan1",
                        json=EV)
        assert r.status_code == 201
        # default channel does not see it
        r = client.get("/events.json?accessKey=SECRET")
        assert r.status_code == 404
        r = client.get("/events.json?accessKey=SECRET&channel=chan1")
        assert r.status_code == 200 and len(r.json()) == 1


class TestFindFilters:
    def _seed(self, client):
        for i, (ev, uid, t) in enumerate([
                ("view", "u1", "2026-01-01T00:00:01.000Z"),
                ("view", "u2", "2026-01-01T00:00:02.000Z"),
                ("buy", "u1", "2026-01-01T00:00:03.000Z")]):
            e = dict(EV, event=ev, entityId=uid, eventTime=t)
            assert client.post("/events.json?accessKey=SECRET",
                               json=e).status_code == 201

    def test_filters(self, client):
        self._seed(client)
        r = client.get("/events.json?accessKey=SECRET")
        assert len(r.json()) == 3
        r = client.get("/events.json?accessKey=SECRET&event=view")
        assert len(r.json()) == 2
        r = client.get("/events.json?accessKey=SECRET&entityType=user"
                       "&entityId=u1")
        assert len(r.json()) == 2
        r = client.get("/events.json?accessKey=SECRET"
                       "&startTime=2026-01-01T00:00:02.000Z")
        assert len(r.json()) == 2
        r = client.get("/events.json?accessKey=SECRET"
                       "&untilTime=2026-01-01T00:00:02.000Z")
        assert len(r.json()) == 1
        r = client.get("/events.json?accessKey=SECRET&limit=2")
        assert len(r.json()) == 2

    def test_reversed_requires_entity(self, client):
        self._seed(client)
        r = client.get("/events.json?accessKey=SECRET&reversed=true")
        assert r.status_code == 400
        r = client.get("/events.json?accessKey=SECRET&reversed=true"
                       "&entityType=user&entityId=u1")
        times = [e["eventTime"] for e in r.json()]
        assert times == sorted(times, reverse=True)


class TestBatch:
    def test_batch_mixed(self, client):
        items = [EV, {"event": "bad"}, dict(EV, entityId="u9")]
        r = client.post("/batch/events.json?accessKey=SECRET", json=items)
        assert r.status_code == 200
        st = [x["status"] for x in r.json()]
        assert st == [201, 400, 201]
        assert "eventId" in r.json()[0] and "message" in r.json()[1]

    def test_batch_cap(self, client):
        items = [EV] * 51
        r = client.post("/batch/events.json?accessKey=SECRET", json=items)
        assert r.status_code == 400
        assert "less than or equal to" in r.json()["message"]


class TestStats:
    def test_stats(self, client):
        client.post("/events.json?accessKey=SECRET", json=EV)
        r = client.get("/stats.json?accessKey=SECRET")
        assert r.status_code == 200
        body = r.json()
        assert body["lifetime"]["status"]["201"] == 1
        ev = body["lifetime"]["events"][0]
        assert ev["event"] == "rate" and ev["count"] == 1


class TestWebhooks:
    def test_segmentio(self, client):
        payload = {"version": "2", "type": "track", "user_id": "u42",
                   "event": "signup",
                   "properties": {"plan": "pro"},
                   "timestamp": "2026-01-02T03:04:05.000Z"}
        r = client.post("/webhooks/segmentio.json?accessKey=SECRET",
                        json=payload)
        assert r.status_code == 201
        r = client.get("/events.json?accessKey=SECRET&event=track")
        e = r.json()[0]
        assert e["entityType"] == "user" and e["entityId"] == "u42"
        assert e["properties"]["event"] == "signup"

    def test_segmentio_no_user(self, client):
        payload = {"version": "2", "type": "track", "event": "x"}
        r = client.post("/webhooks/segmentio.json?accessKey=SECRET",
                        json=payload)
        assert r.status_code == 400

    def test_mailchimp_subscribe(self, client):
        form = {
            "type": "subscribe", "fired_at": "2026-03-26 21:35:57",
            "data[id]": "8a25ff1d98", "data[list_id]": "a6b5da1054",
            "data[email]": "api@mailchimp.com", "data[email_type]": "html",
            "data[merges][EMAIL]": "api@mailchimp.com",
            "data[merges][FNAME]": "MailChimp", "data[merges][LNAME]": "API",
            "data[ip_opt]": "10.20.10.30", "data[ip_signup]": "10.20.10.30",
        }
        r = client.post("/webhooks/mailchimp.form?accessKey=SECRET",
                        data=form)
        assert r.status_code == 201
        e = client.get("/events.json?accessKey=SECRET&event=subscribe"
                       ).json()[0]
        assert e["entityId"] == "8a25ff1d98"
        assert e["targetEntityType"] == "list"
        assert e["properties"]["merges"]["FNAME"] == "MailChimp"

    def test_unknown_connector(self, client):
        r = client.post("/webhooks/nope.json?accessKey=SECRET", json={})
        assert r.status_code == 404

    def test_exampleform(self, client):
        form = {"type": "userActionItem", "event": "view", "userId": "u1",
                "itemId": "i1", "timestamp": "2026-01-02T03:04:05.000Z",
                "context[ip]": "1.2.3.4", "context[prop1]": "2.5",
                "context[prop2]": "value"}
        r = client.post("/webhooks/exampleform.form?accessKey=SECRET",
                        data=form)
        assert r.status_code == 201


class TestPlugins:
    def test_blocker(self, tmp_path, monkeypatch):
        monkeypatch.setenv("PIO_FS_BASEDIR", str(tmp_path))
        monkeypatch.setenv("PIO_STORAGE_SQLITE_PATH",
                           str(tmp_path / "pio.db"))
        storage.reset()
        apps = storage.get_meta_data_apps()
        app_id = apps.insert(App(id=0, name="a"))
        storage.get_meta_data_access_keys().insert(
            AccessKey(key="K", appid=app_id, events=[]))
        storage.get_l_events().init(app_id)

        from predictionio_amd.server.eventserver import (
            EventServerPlugin, create_app,
        )

        class Blocker(EventServerPlugin):
            plugin_name = "noview"
            plugin_type = EventServerPlugin.inputblocker

            def process(self, info):
                if info["event"].event == "view":
                    raise ValueError("view events blocked")

        c = TestClient(create_app(plugins=[Blocker()]))
        ok = dict(EV)
        r = c.post("/events.json?accessKey=K", json=ok)
        assert r.status_code == 201
        r = c.post("/events.json?accessKey=K", json=dict(EV, event="view"))
        assert r.status_code == 403
        storage.reset()


class TestSegmentIOTypes:
    """All six Segment.io message types (SegmentIOConnector.scala:52-92)."""

    def _post(self, client, payload):
        return client.post("/webhooks/segmentio.json?accessKey=SECRET",
                           json=payload)

    def test_identify(self, client):
        r = self._post(client, {"version": "2", "type": "identify",
                                "user_id": "u1",
                                "traits": {"email": "a@b.c"}})
        assert r.status_code == 201
        e = client.get("/events.json?accessKey=SECRET&event=identify"
                       ).json()[0]
        assert e["properties"]["traits"]["email"] == "a@b.c"

    def test_page_screen(self, client):
        for typ in ("page", "screen"):
            r = self._post(client, {"version": "2", "type": typ,
                                    "user_id": "u1", "name": "home"})
            assert r.status_code == 201
            e = client.get(f"/events.json?accessKey=SECRET&event={typ}"
                           ).json()[0]
            assert e["properties"]["name"] == "home"

    def test_alias(self, client):
        r = self._post(client, {"version": "2", "type": "alias",
                                "user_id": "u2", "previous_id": "anon9"})
        assert r.status_code == 201
        e = client.get("/events.json?accessKey=SECRET&event=alias").json()[0]
        assert e["properties"]["previous_id"] == "anon9"

    def test_group(self, client):
        r = self._post(client, {"version": "2", "type": "group",
                                "user_id": "u3", "group_id": "g7",
                                "traits": {"plan": "pro"}})
        assert r.status_code == 201
        e = client.get("/events.json?accessKey=SECRET&event=group").json()[0]
        assert e["properties"]["group_id"] == "g7"

    def test_anonymous_id_fallback(self, client):
        r = self._post(client, {"version": "2", "type": "track",
                                "anonymous_id": "anon1", "event": "x"})
        assert r.status_code == 201
        e = client.get("/events.json?accessKey=SECRET&event=track").json()[0]
        assert e["entityId"] == "anon1"

    def test_context_folded_into_properties(self, client):
        r = self._post(client, {"version": "2", "type": "track",
                                "user_id": "u4", "event": "buy",
                                "context": {"ip": "1.2.3.4"},
                                "properties": {"sku": 9}})
        assert r.status_code == 201
        e = client.get("/events.json?accessKey=SECRET&event=track").json()[0]
        assert e["properties"]["context"]["ip"] == "1.2.3.4"

    def test_missing_version(self, client):
        assert self._post(client, {"type": "track", "user_id": "u",
                                   "event": "x"}).status_code == 400

    def test_unknown_type(self, client):
        assert self._post(client, {"version": "2", "type": "nope",
                                   "user_id": "u"}).status_code == 400


class TestMailChimpTypes:
    def _post(self, client, form):
        return client.post("/webhooks/mailchimp.form?accessKey=SECRET",
                           data=form)

    def test_unsubscribe(self, client):
        r = self._post(client, {
            "type": "unsubscribe", "fired_at": "2026-03-26 21:40:57",
            "data[action]": "unsub", "data[reason]": "manual",
            "data[id]": "8a25ff1d98", "data[list_id]": "a6b5da1054",
            "data[email]": "x@y.z", "data[email_type]": "html",
            "data[merges][EMAIL]": "x@y.z", "data[merges][FNAME]": "X",
            "data[merges][LNAME]": "Y", "data[ip_opt]": "1.1.1.1",
            "data[campaign_id]": "cb398d21d2"})
        assert r.status_code == 201
        e = client.get("/events.json?accessKey=SECRET&event=unsubscribe"
                       ).json()[0]
        assert e["properties"]["action"] == "unsub"

    def test_upemail(self, client):
        r = self._post(client, {
            "type": "upemail", "fired_at": "2026-03-26 22:15:09",
            "data[list_id]": "a6b5da1054", "data[new_id]": "51da8c3259",
            "data[new_email]": "new@x.y", "data[old_email]": "old@x.y"})
        assert r.status_code == 201
        e = client.get("/events.json?accessKey=SECRET&event=upemail"
                       ).json()[0]
        assert e["entityId"] == "51da8c3259"

    def test_cleaned(self, client):
        r = self._post(client, {
            "type": "cleaned", "fired_at": "2026-03-26 22:01:00",
            "data[list_id]": "a6b5da1054", "data[campaign_id]": "c1",
            "data[reason]": "hard", "data[email]": "gone@x.y"})
        assert r.status_code == 201
        e = client.get("/events.json?accessKey=SECRET&event=cleaned"
                       ).json()[0]
        assert e["entityType"] == "list"

    def test_campaign(self, client):
        r = self._post(client, {
            "type": "campaign", "fired_at": "2026-03-26 21:31:21",
            "data[id]": "5aa2102003", "data[subject]": "S",
            "data[status]": "sent", "data[reason]": "",
            "data[list_id]": "a6b5da1054"})
        assert r.status_code == 201
        e = client.get("/events.json?accessKey=SECRET&event=campaign"
                       ).json()[0]
        assert e["entityType"] == "campaign"

    def test_missing_type(self, client):
        assert self._post(client, {"fired_at": "2026-01-01 00:00:00"
                                   }).status_code == 400

    def test_missing_field(self, client):
        assert self._post(client, {"type": "subscribe",
                                   "fired_at": "2026-01-01 00:00:00"
                                   }).status_code == 400
