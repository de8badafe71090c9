from .metrics import METRICS, MetricsRegistry
