"""Engine-template tests on synthetic events (CPU; the GPU numerics of the
underlying kernels are covered by test_gpu_kernels).

Mirrors the role of the reference's template integration scenarios
(tests/pio_tests/scenarios/quickstart_test.py): seed events → train →
predict, asserting behavioral properties (filters honored, fallbacks
taken), not exact scores.
"""

import random

import pytest

from predictionio_amd.data.events import DataMap, Event, utcnow
from predictionio_amd.data.storage.base import App


def _mk_app(storage, name="MyTestApp"):
    app_id = storage.get_meta_data_apps().insert(App(id=0, name=name))
    storage.get_l_events().init(app_id)
    return app_id


def _insert(storage, app_id, event, uid, iid=None, props=None,
            entity_type="user", target_type="item"):
    e = Event(event=event, entity_type=entity_type, entity_id=uid,
              target_entity_type=target_type if iid else None,
              target_entity_id=iid, properties=DataMap(props or {}),
              event_time=utcnow())
    storage.get_l_events().insert(e, app_id)


def _seed_ratings(storage, app_id, n_users=30, n_items=20, seed=1):
    """Two taste clusters: even users like even items, odd like odd."""
    rng = random.Random(seed)
    for u in range(n_users):
        liked = [i for i in range(n_items) if i % 2 == u % 2]
        for i in rng.sample(liked, 6):
            _insert(storage, app_id, "rate", f"u{u}", f"i{i}",
                    {"rating": rng.uniform(4.0, 5.0)})
        disliked = [i for i in range(n_items) if i % 2 != u % 2]
        for i in rng.sample(disliked, 3):
            _insert(storage, app_id, "rate", f"u{u}", f"i{i}",
                    {"rating": rng.uniform(1.0, 2.0)})


class TestRecommendationTemplate:
    def test_train_and_predict(self, mem_storage):
        app_id = _mk_app(mem_storage)
        _seed_ratings(mem_storage, app_id)
        from predictionio_amd.templates.recommendation import (
            ALSAlgorithm, Query, RecommendationEngine,
        )
        from predictionio_amd.controller import EngineParams, Params
        e = RecommendationEngine.apply()
        ep = EngineParams(
            data_source_params=Params({"appName": "MyTestApp"}),
            algorithms_params=[("als", Params(
                {"rank": 8, "numIterations": 8, "lambda": 0.1,
                 "seed": 1}))])
        models = e.train(ep)
        algo = ALSAlgorithm(ep.algorithms_params[0][1])
        r = algo.predict(models[0], Query(user="u2", num=5))
        assert len(r.item_scores) == 5
        # even user should prefer even items
        top = [s.item for s in r.item_scores[:3]]
        evens = sum(1 for it in top if int(it[1:]) % 2 == 0)
        assert evens >= 2

    def test_blacklist(self, mem_storage):
        app_id = _mk_app(mem_storage)
        _seed_ratings(mem_storage, app_id)
        from predictionio_amd.templates.recommendation import (
            ALSAlgorithm, Query, RecommendationEngine,
        )
        from predictionio_amd.controller import EngineParams, Params
        e = RecommendationEngine.apply()
        ep = EngineParams(
            data_source_params=Params({"appName": "MyTestApp"}),
            algorithms_params=[("als", Params(
                {"rank": 8, "numIterations": 5, "seed": 1}))])
        models = e.train(ep)
        algo = ALSAlgorithm(ep.algorithms_params[0][1])
        banned = [f"i{i}" for i in range(0, 20, 2)]
        r = algo.predict(models[0],
                         Query(user="u2", num=5, black_list=banned))
        assert not set(s.item for s in r.item_scores) & set(banned)

    def test_unknown_user(self, mem_storage):
        app_id = _mk_app(mem_storage)
        _seed_ratings(mem_storage, app_id)
        from predictionio_amd.templates.recommendation import (
            ALSAlgorithm, Query, RecommendationEngine,
        )
        from predictionio_amd.controller import EngineParams, Params
        e = RecommendationEngine.apply()
        ep = EngineParams(
            data_source_params=Params({"appName": "MyTestApp"}),
            algorithms_params=[("als", Params(
                {"rank": 8, "numIterations": 3, "seed": 1}))])
        models = e.train(ep)
        algo = ALSAlgorithm(ep.algorithms_params[0][1])
        r = algo.predict(models[0], Query(user="nobody", num=5))
        assert r.item_scores == []

    def test_eval_precision_at_k(self, mem_storage):
        app_id = _mk_app(mem_storage)
        _seed_ratings(mem_storage, app_id)
        from predictionio_amd.templates.recommendation import (
            RecommendationEngine,
        )
        from predictionio_amd.templates.recommendation.evaluation import (
            PrecisionAtK,
        )
        from predictionio_amd.controller import (
            EngineParams, MetricEvaluator, Params,
        )
        e = RecommendationEngine.apply()
        ep = EngineParams(
            data_source_params=Params({"appName": "MyTestApp",
                                       "evalParams": {"kFold": 3,
                                                      "queryNum": 5}}),
            algorithms_params=[("als", Params(
                {"rank": 8, "numIterations": 5, "seed": 1}))])
        res = MetricEvaluator(PrecisionAtK(k=5)).evaluate_base(
            e, e.batch_eval([ep]))
        assert 0.0 <= res.best_score <= 1.0


class TestSimilarProductTemplate:
    def _seed(self, storage, app_id):
        rng = random.Random(2)
        for i in range(20):
            _insert(storage, app_id, "$set", f"i{i}", None,
                    {"categories": ["even" if i % 2 == 0 else "odd"]},
                    entity_type="item", target_type=None)
        for u in range(30):
            for i in (u % 2, u % 2 + 2, u % 2 + 4, u % 2 + 6):
                _insert(storage, app_id, "view", f"u{u}", f"i{i}")

    def test_similar_items(self, mem_storage):
        app_id = _mk_app(mem_storage)
        self._seed(mem_storage, app_id)
        from predictionio_amd.templates.similarproduct import (
            ALSAlgorithm, Query, SimilarProductEngine,
        )
        from predictionio_amd.controller import EngineParams, Params
        e = SimilarProductEngine.apply()
        ep = EngineParams(
            data_source_params=Params({"appName": "MyTestApp"}),
            algorithms_params=[("als", Params(
                {"rank": 8, "numIterations": 8, "seed": 7}))])
        models = e.train(ep)
        algo = ALSAlgorithm(ep.algorithms_params[0][1])
        r = algo.predict(models[0], Query(items=["i0"], num=4))
        assert len(r.item_scores) == 4
        assert "i0" not in [s.item for s in r.item_scores]  # query excluded
        # co-viewed evens should rank high
        top2 = [s.item for s in r.item_scores[:2]]
        assert any(int(t[1:]) % 2 == 0 for t in top2)

    def test_category_filter(self, mem_storage):
        app_id = _mk_app(mem_storage)
        self._seed(mem_storage, app_id)
        from predictionio_amd.templates.similarproduct import (
            ALSAlgorithm, Query, SimilarProductEngine,
        )
        from predictionio_amd.controller import EngineParams, Params
        e = SimilarProductEngine.apply()
        ep = EngineParams(
            data_source_params=Params({"appName": "MyTestApp"}),
            algorithms_params=[("als", Params(
                {"rank": 8, "numIterations": 5, "seed": 7}))])
        models = e.train(ep)
        algo = ALSAlgorithm(ep.algorithms_params[0][1])
        r = algo.predict(models[0],
                         Query(items=["i0"], num=5, categories=["odd"]))
        assert all(int(s.item[1:]) % 2 == 1 for s in r.item_scores)

    def test_cooccurrence(self, mem_storage):
        app_id = _mk_app(mem_storage)
        self._seed(mem_storage, app_id)
        from predictionio_amd.templates.similarproduct import (
            CooccurrenceAlgorithm, Query, SimilarProductEngine,
        )
        from predictionio_amd.controller import EngineParams, Params
        e = SimilarProductEngine.apply()
        ep = EngineParams(
            data_source_params=Params({"appName": "MyTestApp"}),
            algorithms_params=[("cooccurrence", Params({"n": 10}))])
        models = e.train(ep)
        algo = CooccurrenceAlgorithm(Params({"n": 10}))
        r = algo.predict(models[0], Query(items=["i0"], num=3))
        # items co-viewed with i0 are exactly i2, i4, i6 (even users)
        assert set(s.item for s in r.item_scores) <= {"i2", "i4", "i6"}
        assert len(r.item_scores) == 3


class TestECommerceTemplate:
    def _seed(self, storage, app_id):
        for i in range(16):
            _insert(storage, app_id, "$set", f"i{i}", None,
                    {"categories": ["c1"]}, entity_type="item",
                    target_type=None)
        for u in range(20):
            for i in (u % 2, u % 2 + 2, u % 2 + 4):
                _insert(storage, app_id, "view", f"u{u}", f"i{i}")
            _insert(storage, app_id, "buy", f"u{u}", f"i{u % 2}")

    def _engine_params(self, extra=None):
        from predictionio_amd.controller import EngineParams, Params
        p = {"appName": "MyTestApp", "unseenOnly": False,
             "seenEvents": ["buy", "view"], "similarEvents": ["view"],
             "rank": 8, "numIterations": 6, "seed": 11}
        p.update(extra or {})
        return EngineParams(
            data_source_params=Params({"appName": "MyTestApp"}),
            algorithms_params=[("ecomm", Params(p))])

    def test_known_user(self, mem_storage):
        app_id = _mk_app(mem_storage)
        self._seed(mem_storage, app_id)
        from predictionio_amd.templates.ecommercerecommendation import (
            ECommAlgorithm, ECommerceRecommendationEngine, Query,
        )
        e = ECommerceRecommendationEngine.apply()
        ep = self._engine_params()
        models = e.train(ep)
        algo = ECommAlgorithm(ep.algorithms_params[0][1])
        r = algo.predict(models[0], Query(user="u0", num=4))
        assert len(r.item_scores) == 4

    def test_unseen_only_filters_seen(self, mem_storage):
        app_id = _mk_app(mem_storage)
        self._seed(mem_storage, app_id)
        from predictionio_amd.templates.ecommercerecommendation import (
            ECommAlgorithm, ECommerceRecommendationEngine, Query,
        )
        e = ECommerceRecommendationEngine.apply()
        ep = self._engine_params({"unseenOnly": True})
        models = e.train(ep)
        algo = ECommAlgorithm(ep.algorithms_params[0][1])
        r = algo.predict(models[0], Query(user="u0", num=6))
        seen = {"i0", "i2", "i4"}  # u0 viewed/bought these
        assert not set(s.item for s in r.item_scores) & seen

    def test_unavailable_items_constraint(self, mem_storage):
        app_id = _mk_app(mem_storage)
        self._seed(mem_storage, app_id)
        _insert(mem_storage, app_id, "$set", "unavailableItems", None,
                {"items": ["i2", "i4", "i6", "i8"]},
                entity_type="constraint", target_type=None)
        from predictionio_amd.templates.ecommercerecommendation import (
            ECommAlgorithm, ECommerceRecommendationEngine, Query,
        )
        e = ECommerceRecommendationEngine.apply()
        ep = self._engine_params()
        models = e.train(ep)
        algo = ECommAlgorithm(ep.algorithms_params[0][1])
        r = algo.predict(models[0], Query(user="u0", num=6))
        assert not set(s.item for s in r.item_scores) & \
            {"i2", "i4", "i6", "i8"}

    def test_cold_user_falls_back(self, mem_storage):
        app_id = _mk_app(mem_storage)
        self._seed(mem_storage, app_id)
        from predictionio_amd.templates.ecommercerecommendation import (
            ECommAlgorithm, ECommerceRecommendationEngine, Query,
        )
        e = ECommerceRecommendationEngine.apply()
        ep = self._engine_params()
        models = e.train(ep)
        algo = ECommAlgorithm(ep.algorithms_params[0][1])
        # ghost user with no events at all → popularity default;
        # popularity = buy counts and only i0/i1 were ever bought
        r = algo.predict(models[0], Query(user="ghost", num=3))
        assert {s.item for s in r.item_scores} == {"i0", "i1"}

    def test_weighted_items(self, mem_storage):
        app_id = _mk_app(mem_storage)
        self._seed(mem_storage, app_id)
        from predictionio_amd.templates.ecommercerecommendation import (
            ECommAlgorithm, ECommerceRecommendationEngine, Query,
        )
        e = ECommerceRecommendationEngine.apply()
        ep = self._engine_params()
        models = e.train(ep)
        base = ECommAlgorithm(ep.algorithms_params[0][1])
        r0 = base.predict(models[0], Query(user="u0", num=4))
        # crush the top item's score with a tiny weight → it should drop
        top = r0.item_scores[0].item
        ep2 = self._engine_params(
            {"weightedItems": [{"items": [top], "weight": 0.001}]})
        weighted = ECommAlgorithm(ep2.algorithms_params[0][1])
        r1 = weighted.predict(models[0], Query(user="u0", num=4))
        if len(r1.item_scores) > 1 and r0.item_scores[0].score > 0:
            assert r1.item_scores[0].item != top


class TestClassificationTemplate:
    def _seed(self, storage, app_id):
        rng = random.Random(3)
        for u in range(60):
            plan = u % 2
            # plan-0 users: high attr0; plan-1: high attr2
            attr0 = rng.uniform(8, 10) if plan == 0 else rng.uniform(0, 2)
            attr2 = rng.uniform(0, 2) if plan == 0 else rng.uniform(8, 10)
            _insert(storage, app_id, "$set", f"u{u}", None,
                    {"plan": plan, "attr0": attr0, "attr1": 5.0,
                     "attr2": attr2}, target_type=None)

    def test_train_predict(self, mem_storage):
        app_id = _mk_app(mem_storage, "MyApp2")
        self._seed(mem_storage, app_id)
        from predictionio_amd.templates.classification import (
            ClassificationEngine, NaiveBayesAlgorithm, Query,
        )
        from predictionio_amd.controller import EngineParams, Params
        e = ClassificationEngine.apply()
        ep = EngineParams(
            data_source_params=Params({"appName": "MyApp2"}),
            algorithms_params=[("naive", Params({"lambda": 1.0}))])
        models = e.train(ep)
        algo = NaiveBayesAlgorithm(Params({"lambda": 1.0}))
        assert algo.predict(models[0],
                            Query(9.0, 5.0, 1.0)).label == 0.0
        assert algo.predict(models[0],
                            Query(1.0, 5.0, 9.0)).label == 1.0

    def test_accuracy_eval(self, mem_storage):
        app_id = _mk_app(mem_storage, "MyApp2")
        self._seed(mem_storage, app_id)
        from predictionio_amd.templates.classification import engine as ce
        from predictionio_amd.controller import EngineParams, Params
        e = ce.ClassificationEngine.apply()
        ep = EngineParams(
            data_source_params=Params({"appName": "MyApp2", "evalK": 3}),
            algorithms_params=[("naive", Params({"lambda": 1.0}))])
        from predictionio_amd.controller import MetricEvaluator
        res = MetricEvaluator(ce.Accuracy(),
                              [ce.Precision(label=0.0)]).evaluate_base(
            e, e.batch_eval([ep]))
        assert res.best_score > 0.9  # well-separated clusters


class TestRandomForestVariant:
    def test_rf_algorithm(self, mem_storage):
        app_id = _mk_app(mem_storage, "MyApp3")
        TestClassificationTemplate._seed(TestClassificationTemplate(),
                                         mem_storage, app_id)
        from predictionio_amd.templates.classification import engine as ce
        from predictionio_amd.controller import EngineParams, Params
        e = ce.ClassificationEngine.apply()
        ep = EngineParams(
            data_source_params=Params({"appName": "MyApp3"}),
            algorithms_params=[("randomforest",
                                Params({"numTrees": 10, "seed": 0}))])
        models = e.train(ep)
        algo = ce.RandomForestAlgorithm(ep.algorithms_params[0][1])
        assert algo.predict(models[0], ce.Query(9, 5, 1)).label == 0.0
        assert algo.predict(models[0], ce.Query(1, 5, 9)).label == 1.0


class TestRecommendedUserVariant:
    def test_similar_users(self, mem_storage):
        app_id = _mk_app(mem_storage, "FollowApp")
        # two follow communities: even users follow even targets
        for u in range(30):
            for t in (u % 2, u % 2 + 2, u % 2 + 4):
                _insert(mem_storage, app_id, "follow", f"u{u}", f"s{t}",
                        target_type="user")
        from predictionio_amd.templates.similarproduct.engine import (
            RecommendedUserAlgorithm, RecommendedUserEngine, UserQuery,
        )
        from predictionio_amd.controller import EngineParams, Params
        e = RecommendedUserEngine.apply()
        ep = EngineParams(
            data_source_params=Params({"appName": "FollowApp"}),
            algorithms_params=[("als", Params(
                {"rank": 8, "numIterations": 8, "seed": 4}))])
        models = e.train(ep)
        algo = RecommendedUserAlgorithm(ep.algorithms_params[0][1])
        r = algo.predict(models[0], UserQuery(users=["s0"], num=3))
        assert len(r.item_scores) == 3
        assert "s0" not in [s.item for s in r.item_scores]


class TestMultiAlgoEngine:
    def test_als_plus_cooccurrence_combined(self, mem_storage):
        """multi-events-multi-algos variant shape: one engine instance
        running BOTH algorithms, serving combining their predictions."""
        app_id = _mk_app(mem_storage)
        TestSimilarProductTemplate._seed(TestSimilarProductTemplate(),
                                         mem_storage, app_id)
        from predictionio_amd.controller import (
            Engine, EngineParams, Params, Serving,
        )
        from predictionio_amd.templates.similarproduct.engine import (
            ALSAlgorithm, CooccurrenceAlgorithm, DataSource, ItemScore,
            PredictedResult, Preparator, Query,
        )

        class CombinedServing(Serving):
            """Interleave algorithm outputs, dedup by item (the variant's
            custom Serving combines per-algorithm scores)."""

            def serve(self, q, preds):
                seen, out = set(), []
                for rank_pos in range(max(len(p.item_scores)
                                          for p in preds)):
                    for p in preds:
                        if rank_pos < len(p.item_scores):
                            s = p.item_scores[rank_pos]
                            if s.item not in seen:
                                seen.add(s.item)
                                out.append(s)
                return PredictedResult(out[:q.num])

        e = Engine(DataSource, Preparator,
                   {"als": ALSAlgorithm,
                    "cooccurrence": CooccurrenceAlgorithm},
                   CombinedServing)
        ep = EngineParams(
            data_source_params=Params({"appName": "MyTestApp"}),
            algorithms_params=[
                ("als", Params({"rank": 8, "numIterations": 5,
                                "seed": 7})),
                ("cooccurrence", Params({"n": 10}))])
        models = e.train(ep)
        assert len(models) == 2
        algos = e._algorithms(ep)
        serving = e._serving(ep)
        q = Query(items=["i0"], num=5)
        preds = [a.predict(m, q) for a, m in zip(algos, models)]
        combined = serving.serve(q, preds)
        assert 1 <= len(combined.item_scores) <= 5
        assert len({s.item for s in combined.item_scores}) == \
            len(combined.item_scores)  # deduped
