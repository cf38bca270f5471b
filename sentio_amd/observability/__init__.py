from sentio_amd.observability.metrics import MetricsCollector, metrics_collector  # noqa: F401
from sentio_amd.observability.monitoring import PerformanceMonitor, performance_monitor  # noqa: F401
from sentio_amd.observability.tracing import trace_operation  # noqa: F401
