"""Autograd wrapper for the fused NHWC BatchNorm+activation HIP kernels."""

import torch

from .extension import load_extension


class _FusedBNAct(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, training, momentum, eps,
                act, residual, stats, drop_path):
        ext = load_extension()
        x = x.contiguous(memory_format=torch.channels_last)
        if residual is not None:
            residual = residual.contiguous(memory_format=torch.channels_last)
        y, save_mean, save_invstd = ext.bn_act_fwd(
            x, weight, bias, running_mean, running_var, training, momentum, eps, act,
            residual, stats, drop_path)
        ctx.save_for_backward(x, weight, bias, save_mean, save_invstd, drop_path)
        ctx.training = training
        ctx.act = act
        ctx.has_residual = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = load_extension()
        x, weight, bias, save_mean, save_invstd, drop_path = ctx.saved_tensors
        dx, dgamma, dbeta = ext.bn_act_bwd(
            dy, x, weight, bias, save_mean, save_invstd, ctx.training, ctx.act,
            drop_path)
        # the fused "+ residual" passes the upstream grad straight through
        dres = dy if ctx.has_residual else None
        return (dx, dgamma, dbeta, None, None, None, None, None, None, dres,
                None, None)


def fused_bn_act(x, weight, bias, running_mean, running_var, training, momentum, eps,
                 act="silu", residual=None, stats=None, drop_path=None):
    """stats: optional fp32 [buckets, 2, C] per-channel (sum, sumsq) partials
    accumulated by the kernel that PRODUCED x (ops/pwconv.py want_stats path)
    — skips the BN stats read of x. Only meaningful when training.

    drop_path: optional fp32 [B] per-sample keep mask (0 or 1/keep_prob),
    fused as y = act(bn(x)) * drop_path[b] (+ residual) — stochastic depth in
    the same pass (reference drop.py:84-100)."""
    momentum = 0.1 if momentum is None else momentum
    if not training:
        stats = None
    # BN params/stats are fp32 by construction in this framework
    return _FusedBNAct.apply(
        x, weight, bias, running_mean, running_var, bool(training),
        float(momentum), float(eps), act, residual, stats, drop_path)
