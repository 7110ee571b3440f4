from .collector import LatencyWindow, MetricsCollector

__all__ = ["LatencyWindow", "MetricsCollector"]
