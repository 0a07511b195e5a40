from .app import build_app

__all__ = ["build_app"]
