from .server import build_app

__all__ = ["build_app"]
