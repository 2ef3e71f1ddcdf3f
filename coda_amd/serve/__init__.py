from .app import create_app, SelectorSession

__all__ = ["create_app", "SelectorSession"]
