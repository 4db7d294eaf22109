from . import fastio

__all__ = ["fastio"]
