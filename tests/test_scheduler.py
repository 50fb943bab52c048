"""Scheduler behavior tests (step decay + warmup, cosine cycle, factory)."""

import types

import torch

from deepfake_detection_amd.scheduler import (
    CosineLRScheduler,
    StepLRScheduler,
    create_scheduler,
)


def _opt(lr=1.0):
    p = torch.nn.Parameter(torch.zeros(1))
    return torch.optim.SGD([p], lr=lr)


def test_step_lr_decay():
    opt = _opt(1.0)
    sched = StepLRScheduler(opt, decay_t=2, decay_rate=0.5)
    lrs = []
    for epoch in range(6):
        sched.step(epoch)
        lrs.append(opt.param_groups[0]["lr"])
    assert lrs == [1.0, 1.0, 0.5, 0.5, 0.25, 0.25]


def test_step_lr_warmup():
    opt = _opt(1.0)
    sched = StepLRScheduler(opt, decay_t=10, decay_rate=0.9, warmup_t=4, warmup_lr_init=0.2)
    assert abs(opt.param_groups[0]["lr"] - 0.2) < 1e-9  # init to warmup_lr
    sched.step(2)
    assert abs(opt.param_groups[0]["lr"] - 0.6) < 1e-9  # halfway up
    sched.step(4)
    assert abs(opt.param_groups[0]["lr"] - 1.0) < 1e-9


def test_cosine_cycle():
    opt = _opt(1.0)
    sched = CosineLRScheduler(opt, t_initial=10, lr_min=0.0)
    sched.step(0)
    assert abs(opt.param_groups[0]["lr"] - 1.0) < 1e-9
    sched.step(5)
    assert abs(opt.param_groups[0]["lr"] - 0.5) < 1e-6


def test_factory_step():
    args = types.SimpleNamespace(
        epochs=10, sched="step", decay_epochs=2, decay_rate=0.92,
        warmup_lr=1e-4, warmup_epochs=0, min_lr=1e-5, cooldown_epochs=0,
        lr_noise=None, seed=42, eval_metric="loss")
    opt = _opt(0.5)
    sched, num_epochs = create_scheduler(args, opt)
    assert isinstance(sched, StepLRScheduler)
    assert num_epochs == 10


def test_factory_cosine_adds_cooldown():
    args = types.SimpleNamespace(
        epochs=10, sched="cosine", decay_epochs=2, decay_rate=0.1,
        warmup_lr=1e-4, warmup_epochs=0, min_lr=1e-5, cooldown_epochs=3,
        lr_noise=None, seed=42, eval_metric="loss")
    opt = _opt(0.5)
    sched, num_epochs = create_scheduler(args, opt)
    assert num_epochs == 13


def test_lr_noise_deterministic_and_bounded():
    """LR-noise injection (reference scheduler.py:87-105): seeded, applied
    only inside the noise window, and multiplicative around the base LR."""
    import torch

    from deepfake_detection_amd.scheduler import StepLRScheduler

    def make():
        model = torch.nn.Linear(2, 2)
        opt = torch.optim.SGD(model.parameters(), lr=1.0)
        return StepLRScheduler(opt, decay_t=100, decay_rate=1.0,
                               noise_range_t=(2, 100), noise_pct=0.1,
                               noise_seed=42), opt

    s1, o1 = make()
    s2, o2 = make()
    lrs1, lrs2 = [], []
    for t in range(6):
        s1.step(t)
        s2.step(t)
        lrs1.append(o1.param_groups[0]["lr"])
        lrs2.append(o2.param_groups[0]["lr"])
    assert lrs1 == lrs2  # seeded determinism
    assert lrs1[0] == 1.0 and lrs1[1] == 1.0  # before the window: no noise
    assert any(lr != 1.0 for lr in lrs1[2:])  # inside: noise applied
    assert all(0.5 < lr < 1.5 for lr in lrs1)  # bounded (pct=0.1 normal)


def test_cyclic_cosine_reference_semantics():
    """Pin the refactored cycle machinery to hand-computed reference values
    (reference cosine_lr.py): t_mul growth, per-cycle decay, cycle_limit."""
    import math

    opt = _opt(lr=1.0)
    s = CosineLRScheduler(opt, t_initial=4, t_mul=2.0, lr_min=0.1,
                          decay_rate=0.5, cycle_limit=2)
    # t=2: cycle 0 (t_i=4, t_curr=2): 0.1 + 0.5*(1-0.1)*(1+cos(pi/2))
    assert abs(s._get_lr(2)[0] - (0.1 + 0.45 * (1 + math.cos(math.pi / 2)))) < 1e-9
    # t=6: cycle 1 starts at 4, t_i=8, t_curr=2, gamma=0.5
    lo, hi = 0.05, 0.5
    assert abs(s._get_lr(6)[0] - (lo + 0.5 * (hi - lo) * (1 + math.cos(math.pi * 2 / 8)))) < 1e-9
    # t=13: cycle 2 >= cycle_limit -> flat lr_min
    assert abs(s._get_lr(13)[0] - 0.1) < 1e-12


def test_cyclic_tanh_reference_semantics():
    """Tanh specifics: non-prefix warmup ramps toward the in-cycle LR at
    warmup_t; exhausted cycles hold lr_min * decay^cycle_limit."""
    import math

    from deepfake_detection_amd.scheduler import TanhLRScheduler

    opt = _opt(lr=1.0)
    s = TanhLRScheduler(opt, t_initial=10, lb=-6.0, ub=4.0, lr_min=0.01,
                        decay_rate=0.5, warmup_t=2, warmup_lr_init=0.001,
                        cycle_limit=1)
    # warmup target = _get_lr(2) of the decay curve
    tr = 2 / 10
    want2 = 0.01 + 0.5 * (1.0 - 0.01) * (1 - math.tanh(-6.0 * (1 - tr) + 4.0 * tr))
    assert abs(s._get_lr(2)[0] - want2) < 1e-9
    # warmup is linear from warmup_lr_init to that target
    assert abs(s._get_lr(1)[0] - (0.001 + (want2 - 0.001) / 2)) < 1e-9
    # past cycle_limit: floor keeps the final decay applied
    assert abs(s._get_lr(25)[0] - 0.01 * 0.5) < 1e-12
