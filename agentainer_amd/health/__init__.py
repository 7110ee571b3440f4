from .monitor import HealthMonitor

__all__ = ["HealthMonitor"]
