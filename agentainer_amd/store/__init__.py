from .kv import Store

__all__ = ["Store"]
