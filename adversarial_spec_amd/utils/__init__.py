"""Utilities: phase timing, synthetic spec generation."""

from .synth import synthetic_spec
from .timing import PhaseTimer

__all__ = ["synthetic_spec", "PhaseTimer"]
