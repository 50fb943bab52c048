"""1x1 (pointwise) conv on MFMA matrix cores — the production path for the
EfficientNet-family pointwise expand/project convs (reference
dfd/timm/models/efficientnet_blocks.py:277,299 via create_conv2d).

All three passes run on hand-written gfx950 kernels (ops/hip/pwconv.hip):
  fwd        y = x @ w^T          (128x128 block tile, 16x16x32 bf16 MFMA)
  bwd-data   dx = dy @ w          (same kernel, weight transposed once)
  bwd-weight dW = dy^T @ x        (split-M two-stage, fp32 chunk partials)

The forward can additionally emit per-channel sum/sumsq of y into bucketed
fp32 buffers so the following BatchNorm skips its full stats read of y
(`want_stats=True`; consumed by ops.bn_act via functional.bn_act).

Per-shape dispatch: `pw_use_mfma()` consults a baked table measured by
tools/bench_kernels.py on MI355X; unknown shapes default to the MFMA path
when the channel counts are 8-aligned (vectorized LDS staging), MIOpen
otherwise. DFD_AMD_PW_MFMA=0 force-disables (A/B escape hatch).
"""

import os

import torch

from .extension import load_extension

STATS_BUCKETS = 64

# Shapes (C_in, C_out) where MIOpen measured FASTER than the MFMA kernel on
# MI355X (tools/bench_kernels.py --ops pw, 2026-09-14 sweep over every
# B4-299 and deepfake_v4 shape). Only (288,48) and (672,112) were within
# ~25% of MIOpen; keeping them on MFMA preserves the BN-stats epilogue
# (worth more than the gap) and keeps MIOpen out of the step entirely.
# Entries are exceptions, not an allowlist, so unmeasured models still get
# the native path.
_MIOPEN_FASTER: set = set()


def pw_mfma_enabled() -> bool:
    return os.environ.get("DFD_AMD_PW_MFMA", "1") != "0"


def pw_use_mfma(c_in: int, c_out: int) -> bool:
    if not pw_mfma_enabled():
        return False
    if (c_in, c_out) in _MIOPEN_FASTER:
        return False
    return c_in % 8 == 0 and c_out % 8 == 0


def pw_supported(x, weight, stride, padding, dilation, groups) -> bool:
    """Structural eligibility for the MFMA kernels (dtype/layout/1x1)."""
    if groups != 1 or weight.shape[2] != 1 or weight.shape[3] != 1:
        return False
    if any(s != 1 for s in (stride if isinstance(stride, (tuple, list)) else (stride,))):
        return False
    if any(p != 0 for p in (padding if isinstance(padding, (tuple, list)) else (padding,))):
        return False
    if any(d != 1 for d in (dilation if isinstance(dilation, (tuple, list)) else (dilation,))):
        return False
    if x.dtype != torch.bfloat16:
        return False
    return pw_use_mfma(weight.shape[1], weight.shape[0])


def pw_conv2d_fwd(x, weight):
    """bf16 NHWC 1x1 conv forward on matrix cores. No autograd (A/B use)."""
    ext = load_extension()
    x = x.contiguous(memory_format=torch.channels_last)
    return ext.pw_conv2d_fwd_mfma(x, weight)


class _PwConv2d(torch.autograd.Function):
    """1x1 conv with MFMA kernels on all three passes.

    bwd-data reuses the forward kernel: dX[M,K] = dY[M,N] @ W viewed as a
    [K,N]-weighted forward (weight transposed once, tiny). bwd-weight is the
    dedicated split-M kernel (fp32 chunk partials, no atomics).
    """

    @staticmethod
    def forward(ctx, x, weight, stats_out):
        ext = load_extension()
        x = x.contiguous(memory_format=torch.channels_last)
        stats = None
        if stats_out is not None:
            stats = torch.zeros(STATS_BUCKETS, 2, weight.shape[0],
                                device=x.device, dtype=torch.float32)
            stats_out.append(stats)
        y = ext.pw_conv2d_fwd_mfma(x, weight, stats)
        ctx.save_for_backward(x, weight)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = load_extension()
        x, weight = ctx.saved_tensors
        dy = dy.contiguous(memory_format=torch.channels_last)
        N, K = weight.shape[0], weight.shape[1]
        dx = dw = None
        if ctx.needs_input_grad[0]:
            wt = weight.reshape(N, K).t().contiguous().view(K, N, 1, 1)
            dx = ext.pw_conv2d_fwd_mfma(dy, wt, None)
        if ctx.needs_input_grad[1]:
            dw = ext.pw_conv2d_bwd_weight_mfma(dy, x)
        return dx, dw, None


def pw_conv2d(x, weight, bias=None, want_stats=False):
    """Autograd-enabled 1x1 conv on the MFMA kernels (bf16 NHWC).

    Returns y; with want_stats=True (and no bias) the bucketed per-channel
    (sum, sumsq) of y is attached as ``y._dfd_bn_stats = (buckets, M, C)``
    for the following fused BatchNorm to consume (functional.bn_act).
    """
    if weight.dtype != x.dtype:
        weight = weight.to(x.dtype)
    holder = [] if (want_stats and bias is None) else None
    y = _PwConv2d.apply(x, weight, holder)
    if bias is not None:
        y = y + bias.to(y.dtype).view(1, -1, 1, 1)
    elif holder:
        y._dfd_bn_stats = (holder[0],
                           y.shape[0] * y.shape[2] * y.shape[3], y.shape[1])
    return y
