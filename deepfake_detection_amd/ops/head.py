"""Fused classifier + cross-entropy head.

One autograd op computes logits = x @ W^T + b and the label-smoothed
softmax CE loss (reference loss/cross_entropy.py:6-26 semantics; eps=0 is
plain CE) with the fully fused backward (dx, dW, db) — replacing the
GEMM + log_softmax + gather + mean + 4-kernel backward chain of the eager
head. Logits are returned for metrics (accuracy) alongside the loss.
"""

import torch

from .extension import load_extension


class _FusedHeadCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, target, smoothing):
        ext = load_extension()
        loss, logits = ext.head_ce_fwd(x, weight, bias, target, float(smoothing))
        ctx.save_for_backward(x, weight, target, logits)
        ctx.smoothing = float(smoothing)
        ctx.mark_non_differentiable(logits)
        return loss, logits

    @staticmethod
    def backward(ctx, dloss, _dlogits):
        ext = load_extension()
        x, weight, target, logits = ctx.saved_tensors
        dx, dw, db = ext.head_ce_bwd(dloss.float(), logits, x, weight, target,
                                     ctx.smoothing)
        return dx, dw, db, None, None


def fused_head_ce(x, weight, bias, target, smoothing=0.0):
    """(loss, logits) for pooled features x [B, F] against int64 targets."""
    return _FusedHeadCE.apply(x, weight, bias, target, smoothing)
