"""Scheduler factory.

Parity: reference dfd/timm/scheduler/scheduler_factory.py:7-76 — dispatch on
``args.sched``; for the cyclic schedules (cosine/tanh) the returned epoch
count becomes cycle length + cooldown. Builder functions are table-driven
here instead of the reference's if/elif chain.
"""

from .cosine_lr import CosineLRScheduler
from .plateau_lr import PlateauLRScheduler
from .step_lr import StepLRScheduler
from .tanh_lr import TanhLRScheduler


def _noise_kwargs(args, num_epochs):
    window = getattr(args, "lr_noise", None)
    if window is not None:
        if isinstance(window, (list, tuple)):
            window = [n * num_epochs for n in window]
            if len(window) == 1:
                window = window[0]
        else:
            window = window * num_epochs
    return dict(
        noise_range_t=window,
        noise_pct=getattr(args, "lr_noise_pct", 0.67),
        noise_std=getattr(args, "lr_noise_std", 1.0),
        noise_seed=getattr(args, "seed", 42),
    )


def _warmup_kwargs(args):
    return dict(warmup_lr_init=args.warmup_lr, warmup_t=args.warmup_epochs)


def create_scheduler(args, optimizer):
    num_epochs = args.epochs
    noise = _noise_kwargs(args, num_epochs)

    sched = getattr(args, "sched", None)
    if sched in ("cosine", "tanh"):
        cls = CosineLRScheduler if sched == "cosine" else TanhLRScheduler
        kwargs = dict(
            t_initial=num_epochs,
            t_mul=getattr(args, "lr_cycle_mul", 1.0),
            lr_min=args.min_lr,
            cycle_limit=getattr(args, "lr_cycle_limit", 1),
            t_in_epochs=True,
            **_warmup_kwargs(args),
            **noise,
        )
        if sched == "cosine":
            kwargs["decay_rate"] = args.decay_rate
        scheduler = cls(optimizer, **kwargs)
        # cooldown epochs extend the run past the last cycle
        return scheduler, scheduler.get_cycle_length() + args.cooldown_epochs

    if sched == "step":
        return StepLRScheduler(
            optimizer, decay_t=args.decay_epochs, decay_rate=args.decay_rate,
            **_warmup_kwargs(args), **noise), num_epochs

    if sched == "plateau":
        mode = "min" if "loss" in getattr(args, "eval_metric", "") else "max"
        return PlateauLRScheduler(
            optimizer, decay_rate=args.decay_rate, patience_t=args.patience_epochs,
            lr_min=args.min_lr, mode=mode, cooldown_t=0,
            **_warmup_kwargs(args)), num_epochs

    return None, num_epochs
