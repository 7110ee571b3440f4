from .logger import LEVELS, Logger, get_logger, set_global_logger

__all__ = ["LEVELS", "Logger", "get_logger", "set_global_logger"]
