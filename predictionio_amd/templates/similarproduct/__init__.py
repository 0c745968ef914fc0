from predictionio_amd.templates.similarproduct.engine import (  # noqa: F401
    ALSAlgorithm, CooccurrenceAlgorithm, DataSource, Preparator, Query,
    Serving, SimilarProductEngine,
)
