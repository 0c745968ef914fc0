"""Event model, validation, DataMap, aggregation, BiMap tests.

Mirrors the behavior asserted by the reference's data-layer suites
(data/src/test/.../storage/*). Not a port — same semantics, new tests.
"""

from datetime import datetime, timedelta, timezone

import pytest

from predictionio_amd.data.aggregation import aggregate_properties
from predictionio_amd.data.bimap import BiMap
from predictionio_amd.data.events import (
    DataMap, Event, parse_time, validate_event,
)


def ev(**kw):
    defaults = dict(event="view", entity_type="user", entity_id="u1")
    defaults.update(kw)
    return Event(**defaults)


class TestValidation:
    def test_valid_plain_event(self):
        validate_event(ev())

    def test_empty_event_name(self):
        with pytest.raises(ValueError):
            validate_event(ev(event=""))

    def test_empty_entity(self):
        with pytest.raises(ValueError):
            validate_event(ev(entity_id=""))
        with pytest.raises(ValueError):
            validate_event(ev(entity_type=""))

    def test_target_entity_must_pair(self):
        with pytest.raises(ValueError):
            validate_event(ev(target_entity_type="item"))
        with pytest.raises(ValueError):
            validate_event(ev(target_entity_id="i1"))
        validate_event(ev(target_entity_type="item", target_entity_id="i1"))

    def test_unset_requires_properties(self):
        with pytest.raises(ValueError):
            validate_event(ev(event="$unset"))
        validate_event(ev(event="$unset", properties=DataMap({"a": 1})))

    def test_reserved_event_names(self):
        validate_event(ev(event="$set", properties=DataMap({"a": 1})))
        validate_event(ev(event="$delete"))
        with pytest.raises(ValueError):
            validate_event(ev(event="$bogus"))
        with pytest.raises(ValueError):
            validate_event(ev(event="pio_custom"))

    def test_special_event_no_target(self):
        with pytest.raises(ValueError):
            validate_event(ev(event="$set", properties=DataMap({"a": 1}),
                              target_entity_type="item",
                              target_entity_id="i1"))

    def test_reserved_entity_type(self):
        validate_event(ev(entity_type="pio_pr"))  # built-in allowed
        with pytest.raises(ValueError):
            validate_event(ev(entity_type="pio_bogus"))

    def test_reserved_property_prefix(self):
        with pytest.raises(ValueError):
            validate_event(ev(properties=DataMap({"pio_x": 1})))


class TestWireSchema:
    def test_round_trip(self):
        e = ev(target_entity_type="item", target_entity_id="i1",
               properties=DataMap({"rating": 4.5}), tags=["a", "b"],
               pr_id="p1")
        d = e.to_json()
        e2 = Event.from_json(d)
        assert e2.event == "view" and e2.entity_id == "u1"
        assert e2.target_entity_id == "i1"
        assert e2.properties.get("rating", float) == 4.5
        assert e2.tags == ["a", "b"] and e2.pr_id == "p1"
        # wire format is millisecond precision (reference parity)
        assert abs((e2.event_time - e.event_time).total_seconds()) < 0.001

    def test_missing_mandatory(self):
        with pytest.raises(ValueError):
            Event.from_json({"event": "view", "entityType": "user"})

    def test_non_string_mandatory(self):
        with pytest.raises(ValueError):
            Event.from_json({"event": "view", "entityType": "user",
                             "entityId": 4})

    def test_iso8601_time(self):
        e = Event.from_json({"event": "view", "entityType": "u",
                             "entityId": "1",
                             "eventTime": "2026-02-03T04:05:06.789Z"})
        assert e.event_time == datetime(2026, 2, 3, 4, 5, 6, 789000,
                                        tzinfo=timezone.utc)


class TestDataMap:
    def test_typed_get(self):
        dm = DataMap({"a": 1, "b": "x", "c": [1, 2], "d": 2.5, "e": None})
        assert dm.get("a", int) == 1
        assert dm.get("a", float) == 1.0
        assert dm.get("b", str) == "x"
        assert dm.get("c", list) == [1, 2]
        assert dm.get_opt("missing") is None
        assert dm.get_opt("e") is None
        assert dm.get_or_else("missing", 9) == 9
        with pytest.raises(KeyError):
            dm.get("missing")
        with pytest.raises(ValueError):
            dm.get("e")

    def test_union_minus(self):
        a = DataMap({"x": 1, "y": 2})
        b = DataMap({"y": 3, "z": 4})
        assert a.union(b).fields == {"x": 1, "y": 3, "z": 4}
        assert a.minus(["x"]).fields == {"y": 2}


class TestAggregation:
    def test_set_unset_delete(self):
        t0 = datetime(2026, 1, 1, tzinfo=timezone.utc)

        def se(name, props, dt_offset, entity="u1"):
            return Event(event=name, entity_type="user", entity_id=entity,
                         properties=DataMap(props),
                         event_time=t0 + timedelta(minutes=dt_offset))

        events = [
            se("$set", {"a": 1, "b": 2}, 0),
            se("$set", {"b": 3, "c": 4}, 1),
            se("$unset", {"a": None}, 2),
            se("$set", {"x": 9}, 0, entity="u2"),
            se("$delete", {}, 1, entity="u3"),
            se("$set", {"q": 1}, 0, entity="u3"),  # before delete → gone
        ]
        agg = aggregate_properties(events)
        assert agg["u1"].fields == {"b": 3, "c": 4}
        assert agg["u1"].first_updated == t0
        assert agg["u1"].last_updated == t0 + timedelta(minutes=2)
        assert agg["u2"].fields == {"x": 9}
        assert "u3" not in agg

    def test_set_after_delete_restarts(self):
        t0 = datetime(2026, 1, 1, tzinfo=timezone.utc)
        events = [
            Event(event="$set", entity_type="u", entity_id="e",
                  properties=DataMap({"a": 1}), event_time=t0),
            Event(event="$delete", entity_type="u", entity_id="e",
                  event_time=t0 + timedelta(minutes=1)),
            Event(event="$set", entity_type="u", entity_id="e",
                  properties=DataMap({"b": 2}),
                  event_time=t0 + timedelta(minutes=2)),
        ]
        agg = aggregate_properties(events)
        assert agg["e"].fields == {"b": 2}


class TestBiMap:
    def test_bijection(self):
        bm = BiMap.string_int(["b", "a", "b", "c", "a"])
        assert len(bm) == 3
        assert sorted(bm.values()) == [0, 1, 2]
        inv = bm.inverse
        for k in ("a", "b", "c"):
            assert inv[bm[k]] == k

    def test_map_array_and_inverse_table(self):
        bm = BiMap.string_int(["u3", "u1", "u2"])
        arr = bm.map_array(["u1", "u1", "u3"])
        assert list(arr) == [bm["u1"], bm["u1"], bm["u3"]]
        table = bm.inverse_array()
        assert [bm[t] for t in table] == [0, 1, 2]

    def test_unique_values_required(self):
        with pytest.raises(ValueError):
            BiMap({"a": 1, "b": 1})
