from .server import create_app, envelope, run_server

__all__ = ["create_app", "envelope", "run_server"]
