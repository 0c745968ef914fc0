from predictionio_amd.templates.classification.engine import (  # noqa: F401
    ClassificationEngine, DataSource, NaiveBayesAlgorithm, Preparator, Query,
    Serving,
)
