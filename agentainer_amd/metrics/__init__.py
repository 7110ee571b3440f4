from .collector import (DeviceMetricsSampler, LatencyWindow,
                        MetricsCollector)

__all__ = ["DeviceMetricsSampler", "LatencyWindow", "MetricsCollector"]
