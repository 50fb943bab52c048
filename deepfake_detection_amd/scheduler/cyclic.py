"""Shared warmup + restart-cycle machinery for the shaped decay schedules.

The reference implements cosine and tanh decay as two parallel classes with
duplicated cycle arithmetic (reference scheduler/cosine_lr.py, tanh_lr.py);
here the common structure lives once and subclasses provide only the
within-cycle SHAPE: a weight in [0, 1] as a function of the cycle fraction,
multiplying (lr_max - lr_min) above the decayed floor.

Cycle arithmetic (exact parity with the reference):
  * cycle lengths grow geometrically by t_mul (t_i = t_initial * t_mul^i);
  * per-cycle peak and floor decay by decay_rate^i;
  * an optional linear warmup over warmup_t steps, either PREFIXED (cycles
    start after it) or overlapping the first cycle;
  * cycle_limit caps the number of cycles, after which a flat floor holds.
"""

import math

from .scheduler import Scheduler


class CyclicDecayScheduler(Scheduler):
    def __init__(self, optimizer, t_initial: int, t_mul: float = 1.0,
                 lr_min: float = 0.0, decay_rate: float = 1.0, warmup_t=0,
                 warmup_lr_init=0, warmup_prefix=False, cycle_limit=0,
                 t_in_epochs=True, noise_range_t=None, noise_pct=0.67,
                 noise_std=1.0, noise_seed=42, initialize=True):
        super().__init__(
            optimizer, param_group_field="lr", noise_range_t=noise_range_t,
            noise_pct=noise_pct, noise_std=noise_std, noise_seed=noise_seed,
            initialize=initialize)
        assert t_initial > 0
        assert lr_min >= 0
        self.t_initial = t_initial
        self.t_mul = t_mul
        self.lr_min = lr_min
        self.decay_rate = decay_rate
        self.cycle_limit = cycle_limit
        self.warmup_t = warmup_t
        self.warmup_lr_init = warmup_lr_init
        self.warmup_prefix = warmup_prefix
        self.t_in_epochs = t_in_epochs
        if warmup_t:
            targets = self._warmup_targets()
            self.warmup_steps = [(v - warmup_lr_init) / warmup_t for v in targets]
            super().update_groups(warmup_lr_init)
        else:
            self.warmup_steps = [1 for _ in self.base_values]

    # -- subclass hooks ------------------------------------------------------
    def _shape(self, frac: float) -> float:
        """Weight in [0, 1] at cycle fraction `frac` (1 = peak, 0 = floor)."""
        raise NotImplementedError

    def _exhausted_lr(self) -> float:
        """LR once cycle_limit cycles have completed."""
        return self.lr_min

    def _warmup_targets(self):
        """Per-group LR the warmup ramps toward."""
        return self.base_values

    # -- cycle arithmetic ----------------------------------------------------
    def _locate(self, t):
        """(cycle index, position within cycle, cycle length) at tick t."""
        if self.t_mul != 1:
            i = math.floor(
                math.log(1 - t / self.t_initial * (1 - self.t_mul), self.t_mul))
            t_i = self.t_mul ** i * self.t_initial
            t_curr = t - (1 - self.t_mul ** i) / (1 - self.t_mul) * self.t_initial
        else:
            i = t // self.t_initial
            t_i = self.t_initial
            t_curr = t - i * self.t_initial
        return i, t_curr, t_i

    def _get_lr(self, t):
        if t < self.warmup_t:
            return [self.warmup_lr_init + t * s for s in self.warmup_steps]
        if self.warmup_prefix:
            t = t - self.warmup_t
        i, t_curr, t_i = self._locate(t)
        if self.cycle_limit and i >= self.cycle_limit:
            return [self._exhausted_lr() for _ in self.base_values]
        gamma = self.decay_rate ** i
        floor = self.lr_min * gamma
        weight = self._shape(t_curr / t_i)
        return [floor + weight * (v * gamma - floor) for v in self.base_values]

    # -- Scheduler interface -------------------------------------------------
    def get_epoch_values(self, epoch: int):
        return self._get_lr(epoch) if self.t_in_epochs else None

    def get_update_values(self, num_updates: int):
        return self._get_lr(num_updates) if not self.t_in_epochs else None

    def get_cycle_length(self, cycles=0):
        cycles = max(1, cycles or self.cycle_limit)
        if self.t_mul == 1.0:
            return self.t_initial * cycles
        return int(math.floor(
            -self.t_initial * (self.t_mul ** cycles - 1) / (1 - self.t_mul)))
