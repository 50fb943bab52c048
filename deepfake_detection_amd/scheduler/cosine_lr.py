"""Cosine decay with restarts + warmup.

Parity: reference scheduler/cosine_lr.py (SGDR-style schedule); the cycle
machinery lives in CyclicDecayScheduler — this class only supplies the
half-cosine shape.
"""

import math

from .cyclic import CyclicDecayScheduler


class CosineLRScheduler(CyclicDecayScheduler):
    def _shape(self, frac: float) -> float:
        return 0.5 * (1 + math.cos(math.pi * frac))
