from .base import EngineBackend, ModelNotFound
from .echo import EchoEngine

__all__ = ["EngineBackend", "ModelNotFound", "EchoEngine"]
