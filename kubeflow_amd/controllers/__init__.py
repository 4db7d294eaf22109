from .base import Reconciler, ControllerManager, RequeueAfter

__all__ = ["Reconciler", "ControllerManager", "RequeueAfter"]
