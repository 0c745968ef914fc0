from predictionio_amd.templates.recommendation.engine import (  # noqa: F401
    ALSAlgorithm, ALSModel, DataSource, Preparator, Query,
    RecommendationEngine, Serving,
)
