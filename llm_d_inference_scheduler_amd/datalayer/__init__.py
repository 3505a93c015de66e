from .endpoint import Endpoint, EndpointMetadata, Metrics, Role, role_mask  # noqa: F401
from .datastore import Datastore  # noqa: F401
from .attributes import (  # noqa: F401
    PREFIX_CACHE_MATCH_INFO, IN_FLIGHT_LOAD, LATENCY_PREDICTION_INFO,
    PrefixCacheMatchInfo, InFlightLoad, LatencyPredictionInfo,
)
from .runtime import Collector, DataLayerRuntime, DataSource  # noqa: F401
