from .monitor import GpuFaultDetector, HealthMonitor

__all__ = ["GpuFaultDetector", "HealthMonitor"]
