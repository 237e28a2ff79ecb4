"""Per-card (per-GPU) OIM controller."""

from .controller import Controller, ControllerServer

__all__ = ["Controller", "ControllerServer"]
