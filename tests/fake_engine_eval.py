"""Evaluation fixtures resolvable by dotted path for `pio eval` tests."""

from predictionio_amd.controller import Engine, Evaluation, Params, ZeroMetric
from predictionio_amd.controller.metrics import EngineParamsGenerator
from predictionio_amd.controller.engine import EngineParams

from tests.fake_engine import DS0, Prep0, Algo0, Serve0


class ZeroEvaluation(Evaluation):
    def __init__(self):
        super().__init__(engine=Engine(DS0, Prep0, {"": Algo0}, Serve0),
                         metric=ZeroMetric())
        self.engine_params_generator = ZeroGenerator()


class ZeroGenerator(EngineParamsGenerator):
    engine_params_list = [
        EngineParams(algorithms_params=[("", Params())]),
        EngineParams(algorithms_params=[("", Params({"bias": 1}))]),
    ]
