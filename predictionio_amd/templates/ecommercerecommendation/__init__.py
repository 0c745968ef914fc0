from predictionio_amd.templates.ecommercerecommendation.engine import (  # noqa: F401
    DataSource, ECommAlgorithm, ECommerceRecommendationEngine, Preparator,
    Query, Serving,
)
