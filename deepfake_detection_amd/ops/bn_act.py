"""Autograd wrapper for the fused NHWC BatchNorm+activation HIP kernels."""

import torch

from .extension import load_extension


class _FusedBNAct(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, running_mean, running_var, training, momentum, eps,
                act, residual):
        ext = load_extension()
        x = x.contiguous(memory_format=torch.channels_last)
        if residual is not None:
            residual = residual.contiguous(memory_format=torch.channels_last)
            y, save_mean, save_invstd = ext.bn_act_fwd(
                x, weight, bias, running_mean, running_var, training, momentum, eps, act,
                residual)
        else:
            y, save_mean, save_invstd = ext.bn_act_fwd(
                x, weight, bias, running_mean, running_var, training, momentum, eps, act)
        ctx.save_for_backward(x, weight, bias, save_mean, save_invstd)
        ctx.training = training
        ctx.act = act
        ctx.has_residual = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = load_extension()
        x, weight, bias, save_mean, save_invstd = ctx.saved_tensors
        dx, dgamma, dbeta = ext.bn_act_bwd(
            dy, x, weight, bias, save_mean, save_invstd, ctx.training, ctx.act)
        # the fused "+ residual" passes the upstream grad straight through
        dres = dy if ctx.has_residual else None
        return dx, dgamma, dbeta, None, None, None, None, None, None, dres


def fused_bn_act(x, weight, bias, running_mean, running_var, training, momentum, eps,
                 act="silu", residual=None):
    momentum = 0.1 if momentum is None else momentum
    # BN params/stats are fp32 by construction in this framework
    return _FusedBNAct.apply(
        x, weight, bias, running_mean, running_var, bool(training),
        float(momentum), float(eps), act, residual)
