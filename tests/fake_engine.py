"""Purpose-built fake engine suite for controller/workflow tests.

Mirrors the role of the reference's SampleEngine.scala (489 LoC of
deterministic toy DASE components driven by EngineTest.scala) — toy
components with observable behavior, no storage or GPU needed.
"""

from dataclasses import dataclass
from typing import List

from predictionio_amd.controller import (
    Algorithm, DataSource, Engine, EngineFactory, EngineParams, Params,
    PersistentModel, Preparator, SanityCheck, Serving,
)


@dataclass
class TD(SanityCheck):
    """Training data: a list of ints; optionally poisoned for sanity check."""
    xs: List[int]
    poison: bool = False

    def sanity_check(self):
        if self.poison:
            raise ValueError("poisoned training data")


@dataclass
class PD:
    xs: List[int]
    scale: int = 1


@dataclass
class Q:
    x: int


class DS0(DataSource):
    def read_training(self):
        return TD(list(range(self.params.get("n", 4))),
                  poison=self.params.get("poison", False))

    def read_eval(self):
        n = self.params.get("n", 4)
        folds = []
        for k in range(self.params.get("folds", 2)):
            td = TD([x for x in range(n) if x % 2 == k % 2])
            qa = [(Q(x), x * 10) for x in range(3)]
            folds.append((td, {"fold": k}, qa))
        return folds


class Prep0(Preparator):
    def prepare(self, td):
        return PD(td.xs, scale=self.params.get("scale", 1))


class Algo0(Algorithm):
    """Model = sum of scaled data; predict(q) = model + q.x."""

    def train(self, pd):
        return sum(pd.xs) * pd.scale + self.params.get("bias", 0)

    def predict(self, model, q):
        return model + q.x


class Algo1(Algorithm):
    def train(self, pd):
        return len(pd.xs)

    def predict(self, model, q):
        return model * q.x


class AlgoNoPersist(Algo0):
    """persist_model → None ⇒ retrain at deploy."""

    train_count = 0

    def train(self, pd):
        type(self).train_count += 1
        return super().train(pd)

    def persist_model(self, model):
        return None


class SelfSavingModel(PersistentModel):
    """PersistentModel mode: saves itself to a module-level dict."""

    store = {}

    def __init__(self, value):
        self.value = value

    def save(self, instance_id, params):
        type(self).store[instance_id] = self.value
        return True

    @classmethod
    def load(cls, instance_id, params):
        return cls(cls.store[instance_id])


class AlgoPersistent(Algorithm):
    def train(self, pd):
        return SelfSavingModel(sum(pd.xs))

    def predict(self, model, q):
        return model.value + q.x


class Serve0(Serving):
    """Sum predictions; supplement bumps query by `bump` param."""

    def supplement(self, q):
        return Q(q.x + self.params.get("bump", 0))

    def serve(self, q, preds):
        return sum(preds)


class FakeEngineFactory(EngineFactory):
    @classmethod
    def apply(cls):
        return Engine(
            data_source_class=DS0,
            preparator_class=Prep0,
            algorithm_class={"algo0": Algo0, "algo1": Algo1,
                             "nopersist": AlgoNoPersist,
                             "persistent": AlgoPersistent,
                             "": Algo0},
            serving_class=Serve0,
        )


class JsonAlgo(Algorithm):
    """JSON-facing algorithm: dict queries in, dict predictions out
    (the wire contract of /queries.json)."""

    def train(self, pd):
        return sum(pd.xs) * pd.scale

    def predict(self, model, q):
        return {"result": model + q["x"]}


class JsonServing(Serving):
    def serve(self, q, preds):
        return {"result": sum(p["result"] for p in preds)}


class JsonEngineFactory(EngineFactory):
    @classmethod
    def apply(cls):
        return Engine(DS0, Prep0, JsonAlgo, JsonServing)


def make_engine() -> Engine:
    return FakeEngineFactory.apply()


def make_params(algos=None, **kw) -> EngineParams:
    return EngineParams(
        data_source_params=Params(kw.get("ds", {})),
        preparator_params=Params(kw.get("prep", {})),
        algorithms_params=algos or [("algo0", Params())],
        serving_params=Params(kw.get("serving", {})),
    )


RUN_CALLS = []


def fake_main(*args):
    """Target for `pio run` tests (FakeWorkflow.FakeRun parity)."""
    RUN_CALLS.append(args)
    return len(args)
