from .coda import CODA

__all__ = ["CODA"]
