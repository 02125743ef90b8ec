"""CLI layer."""

from .debate import main

__all__ = ["main"]
