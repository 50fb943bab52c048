"""Hyperbolic-tangent decay with restarts.

Parity: reference scheduler/tanh_lr.py — tanh sweep from lb to ub over each
cycle; warmup (non-prefix) ramps toward the in-cycle LR at warmup_t; after
cycle_limit cycles the floor keeps the final cycle's decay applied. Cycle
machinery in CyclicDecayScheduler.
"""

import math

from .cyclic import CyclicDecayScheduler


class TanhLRScheduler(CyclicDecayScheduler):
    def __init__(self, optimizer, t_initial: int, lb: float = -6.0, ub: float = 4.0,
                 **kwargs):
        assert lb < ub
        self.lb = lb
        self.ub = ub
        super().__init__(optimizer, t_initial, **kwargs)

    def _shape(self, frac: float) -> float:
        return 0.5 * (1 - math.tanh(self.lb * (1.0 - frac) + self.ub * frac))

    def _exhausted_lr(self) -> float:
        return self.lr_min * (self.decay_rate ** self.cycle_limit)

    def _warmup_targets(self):
        if self.warmup_prefix:
            return self.base_values
        return self._get_lr(self.warmup_t)
