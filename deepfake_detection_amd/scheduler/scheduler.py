"""Scheduler base: stateless schedules with an epoch/update duality.

Behavioral parity with reference dfd/timm/scheduler/scheduler.py:67-105:
`step(epoch, metric)` applies `get_epoch_values(epoch)`, `step_update(n)`
applies `get_update_values(n)`, and an optional multiplicative LR noise is
injected inside a configured schedule window. The noise draw is a function
of (seed, t) only — rejection-sampled normal truncated at noise_pct, or
uniform in ±noise_pct — so resuming at epoch t reproduces the same LR.
"""

from typing import Any, Dict

import torch


class Scheduler:
    """Base class: subclasses implement `get_epoch_values` and/or
    `get_update_values` as pure functions of the tick index and read
    `self.base_values` (the initial per-group LRs)."""

    def __init__(self, optimizer: torch.optim.Optimizer, param_group_field: str = "lr",
                 noise_range_t=None, noise_type="normal", noise_pct=0.67,
                 noise_std=1.0, noise_seed=None, initialize: bool = True):
        self.optimizer = optimizer
        self.param_group_field = param_group_field
        self._initial_param_group_field = "initial_" + param_group_field
        self._snapshot_initial_values(initialize)
        self.base_values = [
            group[self._initial_param_group_field]
            for group in self.optimizer.param_groups
        ]
        self.metric = None
        self.noise_range_t = noise_range_t
        self.noise_pct = noise_pct
        self.noise_type = noise_type
        self.noise_std = noise_std
        self.noise_seed = 42 if noise_seed is None else noise_seed
        self.update_groups(self.base_values)

    def _snapshot_initial_values(self, initialize: bool) -> None:
        field, initial_field = self.param_group_field, self._initial_param_group_field
        for i, group in enumerate(self.optimizer.param_groups):
            if initialize:
                if field not in group:
                    raise KeyError(f"{field} missing from param_groups[{i}]")
                group.setdefault(initial_field, group[field])
            elif initial_field not in group:
                raise KeyError(f"{initial_field} missing from param_groups[{i}]")

    # -- persistence ---------------------------------------------------------
    def state_dict(self) -> Dict[str, Any]:
        return {k: v for k, v in self.__dict__.items() if k != "optimizer"}

    def load_state_dict(self, state_dict: Dict[str, Any]) -> None:
        self.__dict__.update(state_dict)

    # -- schedule hooks (subclass API) ---------------------------------------
    def get_epoch_values(self, epoch: int):
        return None

    def get_update_values(self, num_updates: int):
        return None

    # -- tick entry points ---------------------------------------------------
    def step(self, epoch: int, metric: float = None) -> None:
        self.metric = metric
        self._apply(self.get_epoch_values(epoch), epoch)

    def step_update(self, num_updates: int, metric: float = None):
        self.metric = metric
        self._apply(self.get_update_values(num_updates), num_updates)

    def _apply(self, values, t) -> None:
        if values is None:
            return
        if self._noise_active(t):
            factor = 1.0 + self._draw_noise(t)
            values = [v * factor for v in values]
        self.update_groups(values)

    def update_groups(self, values):
        if not isinstance(values, (list, tuple)):
            values = [values] * len(self.optimizer.param_groups)
        for group, value in zip(self.optimizer.param_groups, values):
            group[self.param_group_field] = value

    # -- LR noise ------------------------------------------------------------
    def _noise_active(self, t) -> bool:
        window = self.noise_range_t
        if window is None:
            return False
        if isinstance(window, (list, tuple)):
            return window[0] <= t < window[1]
        return t >= window

    def _draw_noise(self, t) -> float:
        """Deterministic per-tick noise: N(0, std) truncated to |x|<pct by
        rejection, or U(-pct, pct)."""
        gen = torch.Generator()
        gen.manual_seed(self.noise_seed + t)
        if self.noise_type == "normal":
            while True:
                sample = torch.randn(1, generator=gen).item() * self.noise_std
                if abs(sample) < self.noise_pct:
                    return sample
        return 2 * (torch.rand(1, generator=gen).item() - 0.5) * self.noise_pct
