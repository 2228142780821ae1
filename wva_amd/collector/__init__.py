from .source import (  # noqa: F401
    MetricResult,
    MetricValue,
    MetricsSource,
    RefreshSpec,
)
from .query_template import QueryList, QueryTemplate  # noqa: F401
from .registry import SourceRegistry  # noqa: F401
from .prometheus_source import PrometheusSource  # noqa: F401
from .replica_metrics import ReplicaMetricsCollector  # noqa: F401
from . import registration  # noqa: F401
