from .requests import COMPLETED, FAILED, PENDING, Request, RequestManager
from .replay import DispatchFn, EngineUnavailable, ReplayWorker

__all__ = [
    "COMPLETED", "FAILED", "PENDING", "Request", "RequestManager",
    "DispatchFn", "EngineUnavailable", "ReplayWorker",
]
