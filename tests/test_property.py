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
                rebuilt[r].append((int(indices[j]),
                                   round(float(values[j]), 4)))
        expect = defaultdict(list)
        for r, c, v in triples:
            expect[r].append((c, round(v, 4)))
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
        got = {(int(r), int(c)): round(float(v), 4)
               for r, c, v in zip(r2, c2, v2)}
        expect = {}
        for r, c, v in triples:  # later entries win
            expect[(r, c)] = round(v, 4)
        assert got == expect
