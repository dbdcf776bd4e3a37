from .metrics import Counter, LatencyTracker, MetricsRegistry  # noqa: F401
from .logging import get_logger  # noqa: F401
from .tracing import trace_range, latency_class  # noqa: F401
