"""Engine train/eval/persistence tests (mirrors core EngineTest.scala:692 in
role: drives Engine.train/eval/prepare_deploy against the fake engine)."""

import pytest

from predictionio_amd.controller import EngineParams, Params, get_engine
from predictionio_amd.workflow import train as train_wf

from tests.fake_engine import (
    AlgoNoPersist, Q, SelfSavingModel, make_engine, make_params,
)


class TestEngineTrain:
    def test_train_single_algo(self):
        e = make_engine()
        models = e.train(make_params(ds={"n": 4}, prep={"scale": 2}))
        assert models == [12]  # sum(0..3)*2

    def test_train_multi_algo(self):
        e = make_engine()
        ep = make_params(algos=[("algo0", Params()), ("algo1", Params()),
                                ("algo0", Params(bias=5))],
                         ds={"n": 3})
        assert e.train(ep) == [3, 3, 8]

    def test_sanity_check_raises(self):
        e = make_engine()
        with pytest.raises(ValueError, match="poisoned"):
            e.train(make_params(ds={"n": 2, "poison": True}))
        # skipped when asked
        e.train(make_params(ds={"n": 2, "poison": True}),
                skip_sanity_check=True)


class TestEngineEval:
    def test_eval_shape_and_serving(self):
        e = make_engine()
        ep = make_params(algos=[("algo0", Params()), ("algo1", Params())],
                         ds={"n": 4, "folds": 2})
        results = e.eval(ep)
        assert len(results) == 2
        eval_info, qpa = results[0]
        assert eval_info == {"fold": 0}
        assert len(qpa) == 3
        # fold0: td=[0,2] → algo0 model=2, algo1 model=2
        # q.x=1: algo0 → 3, algo1 → 2; serving sums → 5
        q, p, a = qpa[1]
        assert (q.x, p, a) == (1, 5, 10)


class TestPersistence:
    def test_auto_persist_round_trip(self):
        e = make_engine()
        ep = make_params(ds={"n": 4})
        models = e.train(ep)
        blob = e.make_serializable_models(ep, "inst1", models)
        out = e.prepare_deploy(ep, "inst1", blob)
        assert out == models

    def test_not_persisted_retrains(self):
        e = make_engine()
        ep = make_params(algos=[("nopersist", Params())], ds={"n": 4})
        AlgoNoPersist.train_count = 0
        models = e.train(ep)
        assert AlgoNoPersist.train_count == 1
        blob = e.make_serializable_models(ep, "inst2", models)
        out = e.prepare_deploy(ep, "inst2", blob)
        assert AlgoNoPersist.train_count == 2  # retrained at deploy
        assert out == models

    def test_persistent_model_manifest(self):
        e = make_engine()
        ep = make_params(algos=[("persistent", Params())], ds={"n": 4})
        models = e.train(ep)
        blob = e.make_serializable_models(ep, "inst3", models)
        assert "inst3" in SelfSavingModel.store
        out = e.prepare_deploy(ep, "inst3", blob)
        assert isinstance(out[0], SelfSavingModel)
        assert out[0].value == 6


class TestEngineJson:
    VARIANT = {
        "id": "variant1",
        "engineFactory": "tests.fake_engine.FakeEngineFactory",
        "datasource": {"params": {"n": 5}},
        "preparator": {"params": {"scale": 3}},
        "algorithms": [
            {"name": "algo0", "params": {"bias": 1}},
            {"name": "algo1", "params": {}},
        ],
        "serving": {"params": {"bump": 2}},
    }

    def test_json_to_engine_params(self):
        e = make_engine()
        ep = e.json_to_engine_params(self.VARIANT)
        assert ep.data_source_params == {"n": 5}
        assert ep.preparator_params == {"scale": 3}
        assert [n for n, _ in ep.algorithms_params] == ["algo0", "algo1"]
        assert ep.algorithms_params[0][1] == {"bias": 1}
        assert ep.serving_params == {"bump": 2}
        models = e.train(ep)
        assert models == [31, 5]  # sum(0..4)*3+1, len

    def test_get_engine_from_factory_path(self):
        e = get_engine("tests.fake_engine.FakeEngineFactory")
        assert e.train(make_params(ds={"n": 2})) == [1]


class TestTrainWorkflow:
    def test_run_train_persists_instance_and_model(self, mem_storage):
        iid = train_wf.run_train_from_variant(TestEngineJson.VARIANT)
        inst = mem_storage.get_meta_data_engine_instances().get(iid)
        assert inst.status == "COMPLETED"
        assert inst.engine_factory == "tests.fake_engine.FakeEngineFactory"
        blob = mem_storage.get_model_data_models().get(iid)
        assert blob is not None
        # round-trip through engine_instance_to_engine_params + prepare_deploy
        e = get_engine(inst.engine_factory)
        ep = e.engine_instance_to_engine_params(inst)
        models = e.prepare_deploy(ep, iid, blob.models)
        assert models == [31, 5]

    def test_failed_train_marks_failed(self, mem_storage):
        variant = dict(TestEngineJson.VARIANT)
        variant["datasource"] = {"params": {"n": 2, "poison": True}}
        with pytest.raises(ValueError):
            train_wf.run_train_from_variant(variant)
        insts = mem_storage.get_meta_data_engine_instances().get_all()
        assert insts and insts[-1].status == "FAILED"


class TestVariantParsing:
    """engine.json → EngineParams edge cases (jValueToEngineParams,
    Engine.scala:355-418 / JsonExtractorSpec analog)."""

    def _engine(self):
        from tests.fake_engine import make_engine
        return make_engine()

    def test_empty_variant_defaults(self):
        ep = self._engine().json_to_engine_params({})
        assert ep.algorithms_params[0][0] == ""
        assert dict(ep.data_source_params) == {}

    def test_bare_params_without_wrapper(self):
        # a stage given as a bare object is treated as its params
        ep = self._engine().json_to_engine_params(
            {"datasource": {"n": 7}})
        assert ep.data_source_params.get("n") == 7

    def test_named_stages_and_multi_algo(self):
        ep = self._engine().json_to_engine_params({
            "datasource": {"name": "dsA", "params": {"n": 3}},
            "serving": {"name": "sB", "params": {"bump": 1}},
            "algorithms": [
                {"name": "algo0", "params": {"bias": 5}},
                {"name": "algo1"},
            ]})
        assert ep.data_source_name == "dsA"
        assert ep.serving_name == "sB"
        assert [n for n, _ in ep.algorithms_params] == ["algo0", "algo1"]
        assert ep.algorithms_params[0][1].get("bias") == 5
        assert dict(ep.algorithms_params[1][1]) == {}

    def test_instance_round_trip(self, mem_storage):
        """EngineParams → stored instance → engine_instance_to_engine_params
        (Engine.scala:420-490)."""
        from predictionio_amd.workflow import train as train_wf
        from tests.fake_engine import make_engine
        variant = {"id": "rt", "engineFactory":
                   "tests.fake_engine.FakeEngineFactory",
                   "datasource": {"params": {"n": 5}},
                   "algorithms": [{"name": "algo0", "params": {"bias": 2}}]}
        iid = train_wf.run_train_from_variant(variant)
        inst = mem_storage.get_meta_data_engine_instances().get(iid)
        ep = make_engine().engine_instance_to_engine_params(inst)
        assert ep.data_source_params.get("n") == 5
        assert ep.algorithms_params == [
            ("algo0", type(ep.data_source_params)({"bias": 2}))] or             ep.algorithms_params[0][0] == "algo0"
        assert ep.algorithms_params[0][1].get("bias") == 2
