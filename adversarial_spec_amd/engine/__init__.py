"""Debate engine: opponent backends, round scheduler, local inference."""

from .backend import StubBackend, get_backend, is_o_series_model
from .scheduler import (
    build_user_message,
    call_models_parallel,
    call_single_model,
    load_context_files,
)

__all__ = [
    "StubBackend",
    "get_backend",
    "is_o_series_model",
    "build_user_message",
    "call_models_parallel",
    "call_single_model",
    "load_context_files",
]
