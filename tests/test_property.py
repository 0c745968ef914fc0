"""Property-based tests (hypothesis) for the data-layer primitives:
aggregation folding, BiMap bijectivity, CSR build and rating dedup are
checked against brute-force references on arbitrary inputs."""

from collections import defaultdict
from datetime import datetime, timedelta, timezone

import torch
from hypothesis import given, settings, strategies as st

from predictionio_amd.data.aggregation import aggregate_properties
from predictionio_amd.data.bimap import BiMap
from predictionio_amd.data.events import DataMap, Event
from predictionio_amd.ops.als import aggregate_ratings, build_csr

T0 = datetime(2026, 1, 1, tzinfo=timezone.utc)

ops = st.sampled_from(["$set", "$unset", "$delete"])
keys = st.sampled_from(["a", "b", "c"])


@st.composite
def event_streams(draw):
    n = draw(st.integers(0, 25))
    evs = []
    for i in range(n):
        op = draw(ops)
        props = {}
        if op in ("$set", "$unset"):
            for k in draw(st.lists(keys, min_size=1, max_size=3,
                                   unique=True)):
                props[k] = draw(st.integers(0, 9))
        evs.append(Event(event=op, entity_type="user",
                         entity_id=draw(st.sampled_from(["e1", "e2"])),
                         properties=DataMap(props),
                         event_time=T0 + timedelta(seconds=i)))
    return evs


class TestAggregationProperties:
    @settings(max_examples=60, deadline=None)
    @given(event_streams())
    def test_matches_naive_fold(self, evs):
        agg = aggregate_properties(evs)
        # brute-force: fold per entity in time order
        state = {}
        for e in sorted(evs, key=lambda e: e.event_time):
            if e.event == "$delete":
                state.pop(e.entity_id, None)
            elif e.event == "$set":
                state.setdefault(e.entity_id, {}).update(
                    e.properties.to_dict())
            elif e.event == "$unset":
                if e.entity_id in state:
                    for k in e.properties.to_dict():
                        state[e.entity_id].pop(k, None)
        expect = {eid: f for eid, f in state.items()}
        assert {eid: pm.to_dict() for eid, pm in agg.items()} == expect


class TestBiMapProperties:
    @settings(max_examples=60, deadline=None)
    @given(st.lists(st.text(min_size=1, max_size=5), min_size=0,
                    max_size=60))
    def test_bijective_contiguous(self, keys_):
        m = BiMap.string_int(keys_)
        distinct = set(keys_)
        assert len(m) == len(distinct)
        assert set(m.values()) == set(range(len(distinct)))
        for k in distinct:
            assert m.get_inverse(m[k]) == k


class TestCSRProperties:
    @settings(max_examples=40, deadline=None)
    @given(st.lists(st.tuples(st.integers(0, 7), st.integers(0, 5),
                              st.floats(0.1, 9, allow_nan=False)),
                    min_size=0, max_size=60))
    def test_csr_roundtrip(self, triples):
        n_rows = 8
        rows = torch.tensor([t[0] for t in triples], dtype=torch.int32)
        cols = torch.tensor([t[1] for t in triples], dtype=torch.int32)
        vals = torch.tensor([t[2] for t in triples], dtype=torch.float32)
        indptr, indices, values = build_csr(rows, cols, vals, n_rows)
        assert indptr[-1] == len(triples)
        rebuilt = defaultdict(list)
        for r in range(n_rows):
            for j in range(int(indptr[r]), int(indptr[r + 1])):
                rebuilt[r].append((int(indices[j]), float(values[j])))
        expect = defaultdict(list)
        for r, c, v in triples:
            # build_csr only permutes: values must round-trip BIT-exactly
            # through their fp32 representation (round(x, 4) straddled
            # fp32 rounding boundaries and was flaky)
            expect[r].append((c, float(torch.tensor(v,
                                                    dtype=torch.float32))))
        for r in range(n_rows):
            assert sorted(rebuilt[r]) == sorted(expect[r])

    @settings(max_examples=40, deadline=None)
    @given(st.lists(st.tuples(st.integers(0, 4), st.integers(0, 4),
                              st.floats(0.1, 9, allow_nan=False)),
                    min_size=0, max_size=40))
    def test_aggregate_sum(self, triples):
        rows = torch.tensor([t[0] for t in triples], dtype=torch.int32)
        cols = torch.tensor([t[1] for t in triples], dtype=torch.int32)
        vals = torch.tensor([t[2] for t in triples], dtype=torch.float32)
        r2, c2, v2 = aggregate_ratings(rows, cols, vals, 5, "sum")
        got = {(int(r), int(c)): float(v)
               for r, c, v in zip(r2, c2, v2)}
        expect = defaultdict(float)
        for r, c, v in triples:
            expect[(r, c)] += v
        assert set(got) == set(expect)
        for k in got:
            assert abs(got[k] - expect[k]) < 1e-4

    @settings(max_examples=40, deadline=None)
    @given(st.lists(st.tuples(st.integers(0, 4), st.integers(0, 4),
                              st.floats(0.1, 9, allow_nan=False)),
                    min_size=0, max_size=40))
    def test_aggregate_latest(self, triples):
        rows = torch.tensor([t[0] for t in triples], dtype=torch.int32)
        cols = torch.tensor([t[1] for t in triples], dtype=torch.int32)
        vals = torch.tensor([t[2] for t in triples], dtype=torch.float32)
        r2, c2, v2 = aggregate_ratings(rows, cols, vals, 5, "latest")
        got = {(int(r), int(c)): float(v)
               for r, c, v in zip(r2, c2, v2)}
        expect = {}
        for r, c, v in triples:  # later entries win; fp32-exact passthrough
            expect[(r, c)] = float(torch.tensor(v, dtype=torch.float32))
        assert got == expect


class TestFindFilterProperties:
    """le.find's 9-dimension filter vs a brute-force list filter on
    arbitrary event sets and filter combos (LEvents.scala:188-200
    semantics: UNSET=any, None=absent, str=equal)."""

    @settings(max_examples=40, deadline=None)
    @given(st.data())
    def test_matches_bruteforce(self, data):
        import itertools
        from predictionio_amd.data import storage as storage_mod
        from predictionio_amd.data.storage.base import UNSET
        from predictionio_amd.data.storage.sqlite import (
            SQLiteClient, SQLiteLEvents,
        )
        le = SQLiteLEvents(SQLiteClient(":memory:"))
        le.init(1)
        n = data.draw(st.integers(0, 30))
        etypes = ["user", "item"]
        enames = ["rate", "buy", "$set"]
        evs = []
        for k in range(n):
            has_tgt = data.draw(st.booleans())
            e = Event(
                event=data.draw(st.sampled_from(enames)),
                entity_type=data.draw(st.sampled_from(etypes)),
                entity_id=data.draw(st.sampled_from(["a", "b", "c"])),
                target_entity_type="item" if has_tgt else None,
                target_entity_id=data.draw(
                    st.sampled_from(["x", "y"])) if has_tgt else None,
                properties=DataMap({}),
                event_time=T0 + timedelta(seconds=data.draw(
                    st.integers(0, 50))))
            le.insert(e, 1)
            evs.append(e)
        kw = {}
        if data.draw(st.booleans()):
            kw["entity_type"] = data.draw(st.sampled_from(etypes))
        if data.draw(st.booleans()):
            kw["entity_id"] = data.draw(st.sampled_from(["a", "b", "z"]))
        if data.draw(st.booleans()):
            kw["event_names"] = data.draw(
                st.lists(st.sampled_from(enames), min_size=1,
                         max_size=2, unique=True))
        if data.draw(st.booleans()):
            kw["start_time"] = T0 + timedelta(
                seconds=data.draw(st.integers(0, 50)))
        if data.draw(st.booleans()):
            kw["until_time"] = T0 + timedelta(
                seconds=data.draw(st.integers(0, 50)))
        tt = data.draw(st.sampled_from(["unset", "none", "item"]))
        if tt != "unset":
            kw["target_entity_type"] = None if tt == "none" else tt
        rev = data.draw(st.booleans())
        got = list(le.find(1, reversed=rev, **kw))

        def keep(e):
            if "entity_type" in kw and e.entity_type != kw["entity_type"]:
                return False
            if "entity_id" in kw and e.entity_id != kw["entity_id"]:
                return False
            if "event_names" in kw and e.event not in kw["event_names"]:
                return False
            if "start_time" in kw and e.event_time < kw["start_time"]:
                return False
            if "until_time" in kw and e.event_time >= kw["until_time"]:
                return False
            if tt == "none" and e.target_entity_type is not None:
                return False
            if tt == "item" and e.target_entity_type != "item":
                return False
            return True

        expect = [e for e in evs if keep(e)]
        assert len(got) == len(expect)
        # ordering: eventTime ascending (descending when reversed)
        times = [e.event_time for e in got]
        assert times == sorted(times, reverse=rev)
        got_keys = sorted((e.event, e.entity_id,
                           e.event_time.isoformat()) for e in got)
        exp_keys = sorted((e.event, e.entity_id,
                           e.event_time.isoformat()) for e in expect)
        assert got_keys == exp_keys


class TestRemoteWireRoundtrip:
    """Property tests for the remote-backend wire marshalling."""

    @given(st.one_of(st.none(), st.text(max_size=20)))
    def test_opt_field_tristate(self, v):
        from predictionio_amd.data.storage.remote import (
            opt_field, opt_unfield,
        )
        from predictionio_amd.data.storage.base import UNSET
        assert opt_unfield(opt_field(v)) == v
        assert opt_unfield(opt_field(UNSET)) is UNSET

    @given(st.datetimes(min_value=datetime(1980, 1, 1),
                        max_value=datetime(2200, 1, 1)))
    def test_ms_roundtrip(self, dt):
        from datetime import timezone

        from predictionio_amd.data.storage.remote import from_ms, ms
        aware = dt.replace(tzinfo=timezone.utc)
        back = from_ms(ms(aware))
        assert abs((back - aware).total_seconds()) < 0.001

    @given(st.text(min_size=1, max_size=30),
           st.text(min_size=1, max_size=30),
           st.dictionaries(st.text(min_size=1, max_size=8).filter(
               lambda s: not s.startswith("$") and not s.startswith("pio_")),
               st.one_of(st.integers(-10**6, 10**6), st.text(max_size=12),
                         st.booleans()), max_size=4))
    def test_engine_instance_roundtrip(self, fac, var, env):
        from predictionio_amd.data.storage.base import (
            EngineInstance, utcnow,
        )
        from predictionio_amd.data.storage.remote import (
            ei_from_dict, ei_to_dict,
        )
        env = {k: str(v) for k, v in env.items()}
        now = utcnow().replace(microsecond=0)
        i = EngineInstance(
            id="x1", status="COMPLETED", start_time=now, end_time=now,
            engine_id="e", engine_version="1", engine_variant=var,
            engine_factory=fac, batch="b", env=env,
            data_source_params='{"a": 1}')
        j = ei_from_dict(ei_to_dict(i))
        assert j.engine_factory == fac and j.engine_variant == var
        assert j.env == env and j.data_source_params == '{"a": 1}'
        assert j.start_time == now
