from .metrics import MetricsEmitter, init_metrics  # noqa: F401
