"""Strategy synthesis: heuristics, optimizer, schedule exploration."""

from .synthesizer import Synthesizer

__all__ = ["Synthesizer"]
